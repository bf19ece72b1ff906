"""Geospatial tests (reference parity: src/test/anovos/test_geospatial.py
— golden-value checks; here hand-computed expectations)."""

import json
import math
import os

import numpy as np
import pandas as pd
import pytest
import torch

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_ingest import geo_auto_detection as gad
from anovos_amd.data_transformer import geo_utils as gu
from anovos_amd.data_transformer import geospatial as geo
from anovos_amd.shared.context import init_context


@pytest.fixture
def ctx():
    return init_context("cpu")


@pytest.fixture
def loc_frame():
    pdf = pd.DataFrame(
        {
            "id": ["a", "a", "b", "b", "b"],
            "latitude": [40.7128, 40.7306, 51.5074, 51.5007, 51.5033],
            "longitude": [-74.0060, -73.9352, -0.1278, -0.1246, -0.1195],
        }
    )
    return AnovosFrame.from_pandas(pdf, device="cpu")


def test_haversine_known_distance():
    # NYC -> London ~ 5570 km
    d = gu.haversine_distance(
        torch.tensor([40.7128]), torch.tensor([-74.0060]),
        torch.tensor([51.5074]), torch.tensor([-0.1278]), unit="km"
    )
    assert float(d) == pytest.approx(5570, rel=0.01)


def test_vincenty_vs_haversine():
    d_h = gu.haversine_distance(torch.tensor([40.0]), torch.tensor([-74.0]),
                                torch.tensor([41.0]), torch.tensor([-73.0]), unit="km")
    d_v = gu.vincenty_distance(torch.tensor([40.0]), torch.tensor([-74.0]),
                               torch.tensor([41.0]), torch.tensor([-73.0]), unit="km")
    assert abs(float(d_h) - float(d_v)) / float(d_v) < 0.01
    # zero distance
    z = gu.vincenty_distance(torch.tensor([40.0]), torch.tensor([-74.0]),
                             torch.tensor([40.0]), torch.tensor([-74.0]))
    assert float(z) == pytest.approx(0.0, abs=1e-6)


def test_geohash_roundtrip():
    lat = torch.tensor([40.7128, 51.5074, -33.8688])
    lon = torch.tensor([-74.0060, -0.1278, 151.2093])
    gh = gu.geohash_encode_int(lat, lon, precision=8)
    strs = gu.geohash_int_to_str(gh, precision=8)
    # canonical geohashes (public): NYC dr5regw3, London gcpvj0du, Sydney r3gx2f77
    assert strs[0].startswith("dr5reg")
    assert strs[1].startswith("gcpvj0")
    assert strs[2].startswith("r3gx2f")
    ints, prec = gu.geohash_str_to_int(strs)
    la2, lo2 = gu.geohash_decode_int(torch.from_numpy(ints), prec)
    assert torch.allclose(la2, lat.to(torch.float64), atol=1e-3)
    assert torch.allclose(lo2, lon.to(torch.float64), atol=1e-3)


def test_point_in_polygon():
    square = [(-1.0, -1.0), (1.0, -1.0), (1.0, 1.0), (-1.0, 1.0)]  # (lon, lat)
    lat = torch.tensor([0.0, 2.0, 0.999, -0.999])
    lon = torch.tensor([0.0, 0.0, 0.999, -1.001])
    inside = gu.point_in_polygon(lat, lon, square)
    assert list(inside.numpy()) == [True, False, True, False]


def test_dms_roundtrip():
    dd = torch.tensor([40.7128, -73.9352])
    d, m, s = gu.dd_to_dms(dd)
    back = gu.dms_to_dd(d, m, s)
    assert torch.allclose(back, dd.to(torch.float64), atol=1e-9)


def test_geo_format_latlon_cartesian_roundtrip(loc_frame):
    odf = geo.geo_format_latlon(loc_frame, ["latitude"], ["longitude"], "dd", "cartesian", result_prefix=["p"])
    assert "p_x" in odf.columns and "p_y" in odf.columns and "p_z" in odf.columns
    odf2 = geo.geo_format_cartesian(odf, ["p_x"], ["p_y"], ["p_z"], "dd", result_prefix=["q"])
    la = odf2.col("q_lat_dd").data.numpy()
    assert la[0] == pytest.approx(40.7128, abs=1e-6)


def test_geo_format_geohash_roundtrip(loc_frame):
    odf = geo.geo_format_latlon(loc_frame, ["latitude"], ["longitude"], "dd", "geohash",
                                result_prefix=["g"], optional_configs={"geohash_precision": 9})
    assert "g_geohash" in odf.columns
    dec = geo.geo_format_geohash(odf, ["g_geohash"], "dd", result_prefix=["d"])
    la = dec.col("d_lat_dd").data.numpy()
    assert la[0] == pytest.approx(40.7128, abs=1e-3)


def test_location_distance(loc_frame):
    odf = loc_frame.with_column("lat2", loc_frame.col("latitude").clone()).with_column(
        "lon2", loc_frame.col("longitude").clone()
    )
    out = geo.location_distance(odf, ["latitude", "longitude"], ["lat2", "lon2"], result_prefix="self")
    d = out.col("self_distance").data.numpy()
    assert np.allclose(d, 0.0, atol=1e-6)


def test_centroid_and_rog(loc_frame):
    cdf = geo.centroid(loc_frame, "latitude", "longitude", id_col="id")
    pdf = cdf.to_pandas().sort_values("id").reset_index(drop=True)
    assert len(pdf) == 2
    # group a: mean of two NYC points
    assert pdf["latitude_centroid"][0] == pytest.approx((40.7128 + 40.7306) / 2, abs=1e-3)
    rdf = geo.rog_calculation(loc_frame, "latitude", "longitude", id_col="id")
    rpdf = rdf.to_pandas()
    assert (rpdf["radius_of_gyration"] >= 0).all()
    g = geo.centroid(loc_frame, "latitude", "longitude")
    assert len(g.to_pandas()) == 1


def test_weighted_centroid(loc_frame):
    wdf = geo.weighted_centroid(loc_frame, "id", "latitude", "longitude")
    pdf = wdf.to_pandas()
    # reference semantics: ONE global weighted centroid per id row
    assert set(pdf.columns) == {"id", "latitude_centroid", "longitude_centroid"}
    assert len(pdf) == 2
    assert pdf["latitude_centroid"].nunique() == 1


def test_location_in_country(ctx, loc_frame):
    odf = geo.location_in_country(ctx, loc_frame, ["latitude"], ["longitude"], "united kingdom", result_prefix=["uk"])
    flags = odf.col("uk_in_united_kingdom").data.numpy()
    assert list(flags) == [0.0, 0.0, 1.0, 1.0, 1.0]


def test_geohash_precision_control(loc_frame):
    odf = geo.geo_format_latlon(loc_frame, ["latitude"], ["longitude"], "dd", "geohash",
                                result_prefix=["g"], optional_configs={"geohash_precision": 9})
    out = geo.geohash_precision_control(odf, ["g_geohash"], output_precision=5)
    c = out.col("g_geohash_precision_5")
    assert all(len(s) == 5 for s in c.dictionary)


def test_ll_gh_cols_detection():
    rng = np.random.default_rng(0)
    pdf = pd.DataFrame(
        {
            "latitude_col": rng.uniform(30, 50, 500),
            "longitude_col": rng.uniform(100, 140, 500),
            "plain_num": rng.normal(5000, 3, 500).round(0),
            "gh": [gad.latlong_to_geo(la, lo, precision=7) for la, lo in zip(rng.uniform(30, 50, 500), rng.uniform(100, 140, 500))],
            "word": ["foo" + str(i % 3) for i in range(500)],
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    lat_cols, long_cols, gh_cols = gad.ll_gh_cols(idf, 10000)
    assert "latitude_col" in lat_cols
    assert "longitude_col" in long_cols
    assert gh_cols == ["gh"]


def test_geo_to_latlong_scalar():
    gh = gad.latlong_to_geo(40.7128, -74.0060, precision=9)
    la = gad.geo_to_latlong(gh, 0)
    lo = gad.geo_to_latlong(gh, 1)
    assert la == pytest.approx(40.7128, abs=1e-3)
    assert lo == pytest.approx(-74.0060, abs=1e-3)


def test_geospatial_analyzer_outputs(tmp_path, loc_frame):
    from anovos_amd.data_analyzer import geospatial_analyzer as ga

    mp = str(tmp_path / "geo")
    ga.descriptive_stats_gen(loc_frame, "latitude", "longitude", None, "id", mp, 3)
    f = os.path.join(mp, "Overall_Summary_1_latitude_longitude.csv")
    assert os.path.exists(f)
    stats = pd.read_csv(f)
    assert int(stats[stats["Stats"] == "Distinct {Lat, Long} Pairs"]["Count"].iloc[0]) == 5
    top = pd.read_csv(os.path.join(mp, "Top_3_Lat_Long_1_latitude_longitude.csv"))
    assert len(top) == 3
    ga.generate_loc_charts_controller(loc_frame, "id", ["latitude"], ["longitude"], None, 100, 1, mp)
    chart = os.path.join(mp, "loc_charts_ll_latitude_longitude")
    assert os.path.exists(chart)
    json.load(open(chart))
