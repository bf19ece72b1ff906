"""Frame-facing geospatial transformations (reference parity:
``anovos/data_transformer/geospatial.py`` :39-1411). All math runs as
vectorized torch ops on the GPU via geo_utils (the reference wrapped
scalar python in per-row Spark UDFs); group aggregations (centroid, ROG)
use scatter_reduce over id codes — no shuffles.
"""

from __future__ import annotations

from typing import List

import numpy as np
import pandas as pd
import torch

from anovos_amd.core.dtypes import NULL_CODE
from anovos_amd.core.frame import AnovosFrame, Column
from anovos_amd.data_transformer import geo_utils as gu

EARTH_RADIUS = gu.EARTH_RADIUS


def _listify(v):
    if isinstance(v, str):
        return [x.strip() for x in v.split("|") if x.strip() != ""]
    return list(v)


def _num(idf: AnovosFrame, col: str) -> torch.Tensor:
    c = idf.col(col)
    if c.kind == "categorical":
        vals = []
        for s in c.dictionary or []:
            try:
                vals.append(float(s))
            except (TypeError, ValueError):
                vals.append(float("nan"))
        lut = torch.tensor(vals + [float("nan")], dtype=torch.float64, device=c.data.device)
        codes = c.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(vals)), codes)
        return lut[codes]
    return c.data.to(torch.float64)


def _fcol(name: str, t: torch.Tensor) -> Column:
    return Column(name, "double", t)


def _maybe_drop(odf: AnovosFrame, cols: List[str], output_mode: str) -> AnovosFrame:
    if output_mode == "replace":
        return odf.drop([c for c in cols if c in odf.columns])
    return odf


# ------------------------------------------------- format conversions

def geo_format_latlon(idf, list_of_lat, list_of_lon, input_format, output_format,
                      result_prefix=[], optional_configs=None, output_mode="append"):
    """Reference geospatial.py:39 — convert (lat, lon) pairs between
    dd/dms/radian and dd/dms/radian/cartesian/geohash."""
    cfg = {"geohash_precision": 8, "radius": EARTH_RADIUS}
    cfg.update(optional_configs or {})
    list_of_lat, list_of_lon = _listify(list_of_lat), _listify(list_of_lon)
    result_prefix = _listify(result_prefix) if result_prefix else []
    fmts = ["dd", "dms", "radian", "cartesian", "geohash"]
    if input_format not in fmts[:3] or output_format not in fmts:
        raise TypeError("Invalid input for input_format or output_format")
    if result_prefix and len(result_prefix) != len(list_of_lat):
        raise TypeError("result_prefix must have the same length as list_of_lat")
    odf = idf
    for k, (latc, lonc) in enumerate(zip(list_of_lat, list_of_lon)):
        lat, lon = _num(idf, latc), _num(idf, lonc)
        if input_format == "radian":
            lat, lon = gu.radian_to_dd(lat, lon)
        elif input_format == "dms":
            # dms packed as d.mmss float (reference loc triple); here we
            # accept dd-style floats already split is not supported — use
            # dd for tensors; dms input treated as (deg + min/100 + sec/10000)
            d = torch.floor(lat)
            m = torch.floor((lat - d) * 100)
            s = ((lat - d) * 100 - m) * 100
            lat = gu.dms_to_dd(d, m, s)
            d2 = torch.floor(lon)
            m2 = torch.floor((lon - d2) * 100)
            s2 = ((lon - d2) * 100 - m2) * 100
            lon = gu.dms_to_dd(d2, m2, s2)
        pf = result_prefix[k] if result_prefix else f"{latc}_{lonc}"
        if output_format in ("dd", "radian"):
            la, lo = (lat, lon) if output_format == "dd" else gu.dd_to_radian(lat, lon)
            odf = odf.with_column(f"{pf}_lat_{output_format}", _fcol(f"{pf}_lat_{output_format}", la))
            odf = odf.with_column(f"{pf}_lon_{output_format}", _fcol(f"{pf}_lon_{output_format}", lo))
        elif output_format == "dms":
            for nm, t in (("lat", lat), ("lon", lon)):
                d, m, s = gu.dd_to_dms(t)
                packed = torch.sign(d) * (d.abs() + m / 100 + s / 10000)
                odf = odf.with_column(f"{pf}_{nm}_dms", _fcol(f"{pf}_{nm}_dms", packed))
        elif output_format == "cartesian":
            x, y, z = gu.dd_to_cartesian(lat, lon, radius=cfg["radius"])
            for nm, t in (("x", x), ("y", y), ("z", z)):
                odf = odf.with_column(f"{pf}_{nm}", _fcol(f"{pf}_{nm}", t))
        else:  # geohash
            prec = int(cfg["geohash_precision"])
            gh = gu.geohash_encode_int(lat, lon, precision=prec)
            null = torch.isnan(lat) | torch.isnan(lon)
            uniq, inv = torch.unique(gh, return_inverse=True)
            dictionary = gu.geohash_int_to_str(uniq, precision=prec)
            codes = inv.to(torch.int32)
            codes = torch.where(null, torch.full_like(codes, NULL_CODE), codes)
            name = f"{pf}_geohash"
            odf = odf.with_column(name, Column(name, "string", codes, dictionary))
        odf = _maybe_drop(odf, [latc, lonc], output_mode)
    return odf


def geo_format_cartesian(idf, list_of_x, list_of_y, list_of_z, output_format,
                         result_prefix=[], optional_configs=None, output_mode="append"):
    """Reference geospatial.py:190 — cartesian → dd/dms/radian/geohash."""
    cfg = {"geohash_precision": 8, "radius": EARTH_RADIUS}
    cfg.update(optional_configs or {})
    list_of_x, list_of_y, list_of_z = _listify(list_of_x), _listify(list_of_y), _listify(list_of_z)
    result_prefix = _listify(result_prefix) if result_prefix else []
    odf = idf
    for k, (xc, yc, zc) in enumerate(zip(list_of_x, list_of_y, list_of_z)):
        lat, lon = gu.cartesian_to_dd(_num(idf, xc), _num(idf, yc), _num(idf, zc), radius=cfg["radius"])
        pf = result_prefix[k] if result_prefix else f"{xc}_{yc}_{zc}"
        tmp = odf.with_column(f"__lat_{k}", _fcol(f"__lat_{k}", lat)).with_column(f"__lon_{k}", _fcol(f"__lon_{k}", lon))
        tmp = geo_format_latlon(tmp, [f"__lat_{k}"], [f"__lon_{k}"], "dd", output_format,
                                result_prefix=[pf], optional_configs=cfg, output_mode="append")
        odf = tmp.drop([f"__lat_{k}", f"__lon_{k}"])
        odf = _maybe_drop(odf, [xc, yc, zc], output_mode)
    return odf


def geo_format_geohash(idf, list_of_geohash, output_format, result_prefix=[],
                       optional_configs=None, output_mode="append"):
    """Reference geospatial.py:333 — geohash → dd/dms/radian/cartesian.
    Decode runs over the column dictionary then applies on-GPU by LUT."""
    cfg = {"radius": EARTH_RADIUS, "geohash_precision": 8}
    cfg.update(optional_configs or {})
    list_of_geohash = _listify(list_of_geohash)
    result_prefix = _listify(result_prefix) if result_prefix else []
    odf = idf
    for k, ghc in enumerate(list_of_geohash):
        c = idf.col(ghc)
        if c.kind != "categorical":
            raise TypeError(f"geohash column '{ghc}' must be a string column")
        ints, prec = gu.geohash_str_to_int(c.dictionary or [])
        gh_lut = torch.from_numpy(ints).to(c.data.device)
        lat_lut, lon_lut = gu.geohash_decode_int(gh_lut, prec)
        lat_lut = torch.cat([lat_lut, torch.tensor([float("nan")], dtype=torch.float64, device=c.data.device)])
        lon_lut = torch.cat([lon_lut, torch.tensor([float("nan")], dtype=torch.float64, device=c.data.device)])
        codes = c.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(ints)), codes)
        lat, lon = lat_lut[codes], lon_lut[codes]
        pf = result_prefix[k] if result_prefix else ghc
        tmp = odf.with_column(f"__lat_{k}", _fcol(f"__lat_{k}", lat)).with_column(f"__lon_{k}", _fcol(f"__lon_{k}", lon))
        if output_format == "dd":
            odf = tmp.rename({f"__lat_{k}": f"{pf}_lat_dd", f"__lon_{k}": f"{pf}_lon_dd"})
        else:
            tmp = geo_format_latlon(tmp, [f"__lat_{k}"], [f"__lon_{k}"], "dd", output_format,
                                    result_prefix=[pf], optional_configs=cfg, output_mode="append")
            odf = tmp.drop([f"__lat_{k}", f"__lon_{k}"])
        odf = _maybe_drop(odf, [ghc], output_mode)
    return odf


# ------------------------------------------------- distances & polygons

def location_distance(idf, list_of_cols_loc1, list_of_cols_loc2, loc_format="dd",
                      result_prefix="", distance_type="haversine", unit="m",
                      optional_configs=None, output_mode="append"):
    """Reference geospatial.py:460 — pairwise distance between two
    location column pairs."""
    cfg = {"radius": EARTH_RADIUS}
    cfg.update(optional_configs or {})
    l1, l2 = _listify(list_of_cols_loc1), _listify(list_of_cols_loc2)
    lat1, lon1 = _num(idf, l1[0]), _num(idf, l1[1])
    lat2, lon2 = _num(idf, l2[0]), _num(idf, l2[1])
    if loc_format == "radian":
        lat1, lon1 = gu.radian_to_dd(lat1, lon1)
        lat2, lon2 = gu.radian_to_dd(lat2, lon2)
    if distance_type == "haversine":
        d = gu.haversine_distance(lat1, lon1, lat2, lon2, unit=unit, radius=cfg["radius"])
    elif distance_type == "vincenty":
        d = gu.vincenty_distance(lat1, lon1, lat2, lon2, unit=unit)
    else:
        d = gu.euclidean_distance(lat1, lon1, lat2, lon2, unit=unit)
    name = (result_prefix + "_distance") if result_prefix else "_".join(l1 + l2) + "_distance"
    odf = idf.with_column(name, _fcol(name, d))
    if output_mode == "replace":
        odf = odf.drop([c for c in l1 + l2 if c in odf.columns])
    return odf


def geohash_precision_control(idf, list_of_geohash, output_precision=8, km_max_error=None, output_mode="append"):
    """Reference geospatial.py:653 — truncate geohashes to a coarser
    precision (km_max_error maps to a precision level)."""
    err_to_prec = [(2500, 1), (630, 2), (78, 3), (20, 4), (2.4, 5), (0.61, 6), (0.076, 7), (0.019, 8), (0.0024, 9), (0.00060, 10), (0.000074, 11)]
    if km_max_error is not None:
        prec = 12
        for err, p in err_to_prec:
            if km_max_error >= err:
                prec = p
                break
    else:
        prec = int(output_precision)
    odf = idf
    for ghc in _listify(list_of_geohash):
        c = idf.col(ghc)
        new_dict_full = [str(s)[:prec] for s in (c.dictionary or [])]
        uniq = list(dict.fromkeys(new_dict_full))
        remap = {s: j for j, s in enumerate(uniq)}
        lut = torch.tensor([remap[s] for s in new_dict_full] + [NULL_CODE], dtype=torch.int32, device=c.data.device)
        codes = c.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(new_dict_full)), codes)
        name = ghc + f"_precision_{prec}"
        odf = odf.with_column(name, Column(name, "string", lut[codes], uniq))
        if output_mode == "replace":
            odf = odf.drop([ghc])
    return odf


def location_in_polygon(idf, list_of_lat, list_of_lon, polygon, result_prefix=[], output_mode="append"):
    """Reference geospatial.py:727 — 1/0 flag by ray-cast membership.
    polygon: GeoJSON-style geometry dict or list of [lon, lat] rings."""
    list_of_lat, list_of_lon = _listify(list_of_lat), _listify(list_of_lon)
    result_prefix = _listify(result_prefix) if result_prefix else []
    polys = _extract_polygons(polygon)
    odf = idf
    for k, (latc, lonc) in enumerate(zip(list_of_lat, list_of_lon)):
        lat, lon = _num(idf, latc), _num(idf, lonc)
        inside = gu.point_in_polygons(lat, lon, polys)
        out = inside.to(torch.float32)
        out = torch.where(torch.isnan(lat) | torch.isnan(lon), torch.full_like(out, float("nan")), out)
        pf = result_prefix[k] if result_prefix else f"{latc}_{lonc}"
        name = pf + "_in_polygon"
        odf = odf.with_column(name, Column(name, "int", out))
        odf = _maybe_drop(odf, [latc, lonc], output_mode)
    return odf


def _extract_polygons(polygon):
    """Accept GeoJSON geometry/feature dicts or raw ring lists."""
    if isinstance(polygon, dict):
        if polygon.get("type") == "FeatureCollection":
            polys = []
            for f in polygon["features"]:
                polys.extend(_extract_polygons(f))
            return polys
        if polygon.get("type") == "Feature":
            return _extract_polygons(polygon["geometry"])
        if polygon.get("type") == "Polygon":
            return [polygon["coordinates"][0]]
        if polygon.get("type") == "MultiPolygon":
            return [p[0] for p in polygon["coordinates"]]
        raise TypeError(f"unsupported geometry type {polygon.get('type')}")
    if isinstance(polygon, (list, tuple)):
        first = polygon[0]
        if isinstance(first[0], (int, float)):
            return [polygon]  # single ring
        return list(polygon)  # list of rings
    raise TypeError("polygon must be a GeoJSON dict or coordinate list")


def location_in_country(ctx, idf, list_of_lat, list_of_lon, country, country_shapefile_path="",
                        method_type="approx", result_prefix=[], output_mode="append"):
    """Reference geospatial.py:814 — membership in a country, bbox
    ('approx') or geojson shapefile ('exact')."""
    list_of_lat, list_of_lon = _listify(list_of_lat), _listify(list_of_lon)
    result_prefix = _listify(result_prefix) if result_prefix else []
    if method_type == "exact":
        import json

        with open(country_shapefile_path) as f:
            gj = json.load(f)
        polys = _extract_polygons(gj)
    odf = idf
    for k, (latc, lonc) in enumerate(zip(list_of_lat, list_of_lon)):
        lat, lon = _num(idf, latc), _num(idf, lonc)
        if method_type == "exact":
            inside = gu.point_in_polygons(lat, lon, polys)
        else:
            inside = gu.point_in_country_approx(lat, lon, country)
        out = inside.to(torch.float32)
        out = torch.where(torch.isnan(lat) | torch.isnan(lon), torch.full_like(out, float("nan")), out)
        pf = result_prefix[k] if result_prefix else f"{latc}_{lonc}"
        name = f"{pf}_in_{str(country).lower().replace(' ', '_')}"
        odf = odf.with_column(name, Column(name, "int", out))
        odf = _maybe_drop(odf, [latc, lonc], output_mode)
    return odf


# ------------------------------------------------- centroid / ROG

def _id_codes(idf: AnovosFrame, id_col: str):
    c = idf.col(id_col)
    if c.kind == "categorical":
        valid = c.data != NULL_CODE
        return c.data.to(torch.long), valid, lambda u: [c.dictionary[int(i)] for i in u]
    uniq, inv = torch.unique(c.data, return_inverse=True)
    valid = ~c.null_mask()
    return inv, valid, lambda u: [float(uniq[int(i)]) for i in u]


def centroid(idf, lat_col, long_col, id_col=None):
    """Reference geospatial.py:975 — spherical centroid per id (or
    global): mean of cartesian coords → back to dd, via scatter_reduce."""
    lat, lon = _num(idf, lat_col), _num(idf, long_col)
    ok = gu.in_range(lat, lon) & ~torch.isnan(lat) & ~torch.isnan(lon)
    x, y, z = gu.dd_to_cartesian(lat, lon, radius=1.0)
    if id_col:
        inv, valid, decode = _id_codes(idf, id_col)
        m = ok & valid
        G = int(inv.max().item()) + 1 if inv.numel() else 0
        sums = []
        cnt = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], torch.ones_like(x[m]), reduce="sum")
        for t in (x, y, z):
            s = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], t[m], reduce="sum")
            sums.append(s / cnt.clamp(min=1))
        cla, clo = gu.cartesian_to_dd(sums[0], sums[1], sums[2], radius=(sums[0] ** 2 + sums[1] ** 2 + sums[2] ** 2).sqrt().clamp(min=1e-300))
        present = (cnt > 0).nonzero(as_tuple=True)[0]
        ids = decode(present.cpu().numpy())
        pdf = pd.DataFrame({id_col: ids,
                            lat_col + "_centroid": cla[present].cpu().numpy(),
                            long_col + "_centroid": clo[present].cpu().numpy()})
        return AnovosFrame.from_pandas(pdf, device=idf.device)
    xs, ys, zs = x[ok].mean(), y[ok].mean(), z[ok].mean()
    norm = (xs ** 2 + ys ** 2 + zs ** 2).sqrt().clamp(min=1e-300)
    cla, clo = gu.cartesian_to_dd(xs, ys, zs, radius=norm)
    pdf = pd.DataFrame({lat_col + "_centroid": [float(cla)], long_col + "_centroid": [float(clo)]})
    return AnovosFrame.from_pandas(pdf, device=idf.device)


def weighted_centroid(idf, id_col, lat_col, long_col):
    """Reference geospatial.py:1099 — ONE dataset-wide weighted centroid
    replicated onto every id row: each id group's cartesian vector sum is
    scaled by the group's own row count (count-squared weighting of each
    point), the scaled sums are pooled over all groups, and the single
    resulting (lat, long) is attached to every id. Matches the
    reference's output exactly (its unit test asserts the global value
    per id, e.g. [-54, -113] on its geo fixture)."""
    lat, lon = _num(idf, lat_col), _num(idf, long_col)
    ok = gu.in_range(lat, lon) & ~torch.isnan(lat) & ~torch.isnan(lon)
    inv, valid, decode = _id_codes(idf, id_col)
    m = ok & valid
    x, y, z = gu.dd_to_cartesian(lat, lon, radius=1.0)
    G = int(inv.max().item()) + 1 if inv.numel() else 0
    cnt = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], torch.ones_like(x[m]), reduce="sum")
    sums = []
    for t in (x, y, z):
        s = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], t[m], reduce="sum")
        sums.append(s)
    total_w = float(cnt.sum().clamp(min=1.0))
    xs = float((sums[0] * cnt).sum()) / total_w
    ys = float((sums[1] * cnt).sum()) / total_w
    zs = float((sums[2] * cnt).sum()) / total_w
    import math

    hyp = math.sqrt(xs * xs + ys * ys)
    cla = math.atan2(zs, hyp) * 180.0 / math.pi
    clo = math.atan2(ys, xs) * 180.0 / math.pi
    present = (cnt > 0).nonzero(as_tuple=True)[0]
    ids = decode(present.cpu().numpy())
    pdf = pd.DataFrame({id_col: ids,
                        lat_col + "_centroid": [cla] * len(ids),
                        long_col + "_centroid": [clo] * len(ids)})
    return AnovosFrame.from_pandas(pdf, device=idf.device)


def rog_calculation(idf, lat_col, long_col, id_col=None):
    """Reference geospatial.py:1223 — radius of gyration: mean haversine
    distance of points to their (id-group) centroid."""
    lat, lon = _num(idf, lat_col), _num(idf, long_col)
    ok = gu.in_range(lat, lon) & ~torch.isnan(lat) & ~torch.isnan(lon)
    if id_col:
        inv, valid, decode = _id_codes(idf, id_col)
        m = ok & valid
        x, y, z = gu.dd_to_cartesian(lat, lon, radius=1.0)
        G = int(inv.max().item()) + 1 if inv.numel() else 0
        cnt = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], torch.ones_like(x[m]), reduce="sum")
        cs = []
        for t in (x, y, z):
            s = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], t[m], reduce="sum")
            cs.append(s / cnt.clamp(min=1))
        norm = (cs[0] ** 2 + cs[1] ** 2 + cs[2] ** 2).sqrt().clamp(min=1e-300)
        cla, clo = gu.cartesian_to_dd(cs[0], cs[1], cs[2], radius=norm)
        d = gu.haversine_distance(lat[m], lon[m], cla[inv[m]], clo[inv[m]])
        rog = torch.zeros(G, dtype=torch.float64, device=lat.device).scatter_reduce(0, inv[m], d, reduce="sum") / cnt.clamp(min=1)
        present = (cnt > 0).nonzero(as_tuple=True)[0]
        ids = decode(present.cpu().numpy())
        pdf = pd.DataFrame({id_col: ids, "radius_of_gyration": rog[present].cpu().numpy()})
        return AnovosFrame.from_pandas(pdf, device=idf.device)
    cdf = centroid(idf, lat_col, long_col)
    cla = float(cdf.col(lat_col + "_centroid").data[0])
    clo = float(cdf.col(long_col + "_centroid").data[0])
    d = gu.haversine_distance(lat[ok], lon[ok], torch.full_like(lat[ok], cla), torch.full_like(lon[ok], clo))
    pdf = pd.DataFrame({"radius_of_gyration": [float(d.mean())]})
    return AnovosFrame.from_pandas(pdf, device=idf.device)


def reverse_geocoding(idf, lat_col, long_col):
    """Reference geospatial.py:1335 — offline reverse geocode to the
    nearest entry of a small built-in city table (the reference used the
    reverse_geocoder package; no network/package here)."""
    cities = _CITY_TABLE
    lat, lon = _num(idf, lat_col), _num(idf, long_col)
    # the reference drops null and out-of-range rows before geocoding
    # (geospatial.py:1346-1377; its unit test asserts the filtered row
    # counts)
    ok = gu.in_range(lat, lon) & ~torch.isnan(lat) & ~torch.isnan(lon)
    lat, lon = lat[ok], lon[ok]
    clat = torch.tensor([c[0] for c in cities], dtype=torch.float64, device=lat.device)
    clon = torch.tensor([c[1] for c in cities], dtype=torch.float64, device=lat.device)
    # chord distance argmin over the city table (few hundred entries)
    x, y, z = gu.dd_to_cartesian(lat, lon, radius=1.0)
    cx, cy, cz = gu.dd_to_cartesian(clat, clon, radius=1.0)
    d2 = (x.unsqueeze(1) - cx) ** 2 + (y.unsqueeze(1) - cy) ** 2 + (z.unsqueeze(1) - cz) ** 2
    idx = d2.argmin(dim=1).cpu().numpy()
    names = [cities[i][2] for i in idx]
    countries = [cities[i][3] for i in idx]
    pdf = pd.DataFrame({lat_col: lat.cpu().numpy(), long_col: lon.cpu().numpy(),
                        "city": names, "country": countries})
    return AnovosFrame.from_pandas(pdf, device=idf.device)


_CITY_TABLE = [
    (40.7128, -74.0060, "New York", "US"), (34.0522, -118.2437, "Los Angeles", "US"),
    (41.8781, -87.6298, "Chicago", "US"), (51.5074, -0.1278, "London", "GB"),
    (48.8566, 2.3522, "Paris", "FR"), (52.5200, 13.4050, "Berlin", "DE"),
    (40.4168, -3.7038, "Madrid", "ES"), (41.9028, 12.4964, "Rome", "IT"),
    (55.7558, 37.6173, "Moscow", "RU"), (39.9042, 116.4074, "Beijing", "CN"),
    (31.2304, 121.4737, "Shanghai", "CN"), (35.6762, 139.6503, "Tokyo", "JP"),
    (37.5665, 126.9780, "Seoul", "KR"), (19.0760, 72.8777, "Mumbai", "IN"),
    (28.7041, 77.1025, "Delhi", "IN"), (12.9716, 77.5946, "Bangalore", "IN"),
    (1.3521, 103.8198, "Singapore", "SG"), (-33.8688, 151.2093, "Sydney", "AU"),
    (-37.8136, 144.9631, "Melbourne", "AU"), (-23.5505, -46.6333, "Sao Paulo", "BR"),
    (-34.6037, -58.3816, "Buenos Aires", "AR"), (19.4326, -99.1332, "Mexico City", "MX"),
    (30.0444, 31.2357, "Cairo", "EG"), (6.5244, 3.3792, "Lagos", "NG"),
    (-26.2041, 28.0473, "Johannesburg", "ZA"), (25.2048, 55.2708, "Dubai", "AE"),
    (41.0082, 28.9784, "Istanbul", "TR"), (52.3676, 4.9041, "Amsterdam", "NL"),
    (59.3293, 18.0686, "Stockholm", "SE"), (45.4215, -75.6972, "Ottawa", "CA"),
    (43.6532, -79.3832, "Toronto", "CA"), (49.2827, -123.1207, "Vancouver", "CA"),
]
