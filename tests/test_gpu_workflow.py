"""GPU end-to-end: the full income YAML pipeline on cuda:0, HIP kernels
loaded (fails loudly if the extension is missing on a GPU box)."""

import os
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@requires_gpu
def test_income_workflow_on_gpu(tmp_path, monkeypatch):
    from anovos_amd.ops import backend

    assert backend.require_hip() is not None, "HIP extension must load on a GPU box"

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    monkeypatch.chdir(tmp_path)
    sys.path.insert(0, os.path.join(repo, "tools"))
    import make_income_data

    df = make_income_data.make(20000)
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    os.makedirs("data/income_dataset/parquet", exist_ok=True)
    os.makedirs("data/income_dataset/join", exist_ok=True)
    os.makedirs("data/income_dataset/source/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    df.to_parquet("data/income_dataset/parquet/part-00000.parquet")
    from anovos_amd.core.avro_codec import write_avro

    join_df = df[["ifa", "age", "workclass"]].rename(columns={"age": "dupl_age", "workclass": "dupl_workclass"})
    write_avro(join_df, "data/income_dataset/join/part-00000.avro")
    src = make_income_data.make(20000, seed=12)
    src["age"] = src["age"] * 1.05
    src.to_csv("data/income_dataset/source/csv/part-00000.csv", index=False)

    from anovos_amd import workflow

    out = workflow.run(os.path.join(repo, "config", "configs.yaml"), device="cuda:0")
    assert out is not None
    assert out.device.type == "cuda"
    assert os.path.exists("report_stats/ml_anovos_report.html")
    assert os.path.getsize("report_stats/ml_anovos_report.html") > 1_000_000


@requires_gpu
def test_datetime_ops_on_gpu():
    """Datetime civil-decompose arithmetic stays on-device."""
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_transformer import datetime as adt

    pdf = pd.DataFrame({"ts": pd.to_datetime(["2020-01-01 00:00:00", "2020-02-29 13:45:10", "2021-12-31 23:59:59"])})
    idf = AnovosFrame.from_pandas(pdf, device="cuda:0")
    odf = adt.timeUnits_extraction(idf, ["ts"], "all")
    y = odf.col("ts_year")
    assert y.data.is_cuda
    assert y.data.cpu().tolist() == [2020.0, 2020.0, 2021.0]
    b = adt.end_of_month(idf, ["ts"])
    assert b.col("ts_monthEnd").data.is_cuda


@requires_gpu
def test_geospatial_ops_on_gpu():
    import pandas as pd
    import torch

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_transformer import geospatial as geo

    pdf = pd.DataFrame({"lat": [40.7128, 51.5074], "lon": [-74.0060, -0.1278]})
    idf = AnovosFrame.from_pandas(pdf, device="cuda:0")
    odf = geo.geo_format_latlon(idf, ["lat"], ["lon"], "dd", "geohash", result_prefix=["g"])
    gh = odf.col("g_geohash")
    assert gh.data.is_cuda
    assert gh.dictionary[int(gh.data[0])].startswith("dr5reg")
    d = geo.location_distance(idf, ["lat", "lon"], ["lat", "lon"], result_prefix="self")
    assert float(d.col("self_distance").data.abs().max()) < 1e-6


@requires_gpu
def test_edge_cases_on_gpu():
    """Degenerate columns through the GPU kernel paths (constant,
    all-null, inf, null-only categorical) — same assertions as the CPU
    edge suite."""
    from tests.test_edge_cases import edge_frame
    from anovos_amd.data_analyzer import quality_checker as qc
    from anovos_amd.data_analyzer import stats_generator as sg
    from anovos_amd.data_transformer import transformers as T
    from anovos_amd.shared.context import AnovosContext

    ctx = AnovosContext("cuda:0")
    idf = edge_frame("cuda:0")
    m = sg.measures_of_counts(ctx, idf).set_index("attribute")
    assert int(m.loc["all_null", "missing_count"]) == 5000
    assert int(m.loc["null_cat", "missing_count"]) == 5000
    d = sg.measures_of_dispersion(ctx, idf).set_index("attribute")
    assert float(d.loc["constant", "stddev"]) == pytest.approx(0.0, abs=1e-9)
    p = sg.measures_of_percentiles(ctx, idf, ["normal", "constant"]).set_index("attribute")
    assert float(p.loc["constant", "50%"]) == pytest.approx(3.14, rel=1e-6)
    odf = T.attribute_binning(ctx, idf, ["normal", "constant"], bin_size=5, output_mode="append")
    assert "constant_binned" in odf.columns
    _, nr = qc.nullRows_detection(ctx, idf, treatment=False)
    assert int(nr["row_count"].sum()) == 5000


@requires_gpu
def test_geospatial_config_on_gpu(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "tools"))
    import make_income_data as mid

    df = mid.add_geo_cols(mid.make(15000))
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    from anovos_amd import workflow

    out = workflow.run(os.path.join(repo, "config", "configs_geospatial.yaml"), device="cuda:0")
    assert out.device.type == "cuda"
    assert "radius_of_gyration" in out.columns


@requires_gpu
def test_time_series_config_on_gpu(tmp_path, monkeypatch):
    monkeypatch.chdir(tmp_path)
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(repo, "tools"))
    import make_income_data as mid

    df = mid.add_ts_cols(mid.make(8000))
    os.makedirs("data/income_dataset/csv", exist_ok=True)
    df.to_csv("data/income_dataset/csv/part-00000.csv", index=False)
    for k in range(12):
        snap = mid.make(1500, seed=100 + k)
        snap["age"] = snap["age"] * (1 + 0.01 * k)
        d = f"data/income_dataset/snapshot{k + 1:02d}/csv"
        os.makedirs(d, exist_ok=True)
        snap.to_csv(os.path.join(d, "part-00000.csv"), index=False)
    from anovos_amd import workflow

    out = workflow.run(os.path.join(repo, "config", "configs_time_series.yaml"), device="cuda:0")
    assert out.device.type == "cuda"
    assert os.path.exists("report_stats/stability_index.csv")


@requires_gpu
def test_invariants_on_gpu():
    """Randomized invariant checks on the GPU sketch paths."""
    from tests.test_invariants import random_frame
    from anovos_amd.data_analyzer import stats_generator as sg
    from anovos_amd.shared.context import AnovosContext

    ctx = AnovosContext("cuda:0")
    for seed in (21, 22):
        idf = random_frame(seed, n=2_000_000).to_device("cuda:0")
        total = idf.count()
        m = sg.measures_of_counts(ctx, idf).set_index("attribute")
        for attr in m.index:
            assert int(m.loc[attr, "fill_count"]) + int(m.loc[attr, "missing_count"]) == total
        p = sg.measures_of_percentiles(ctx, idf, ["a", "b", "c"]).set_index("attribute")
        qs = ["1%", "5%", "10%", "25%", "50%", "75%", "90%", "95%", "99%"]
        for attr in p.index:
            vals = [float(p.loc[attr, q]) for q in qs]
            vals = [v for v in vals if v == v]
            assert vals == sorted(vals), (attr, vals)
