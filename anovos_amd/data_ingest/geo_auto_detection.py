"""Auto-detect latitude / longitude / geohash columns (reference parity:
``anovos/data_ingest/geo_auto_detection.py`` :22-298).

MI355X-native: the reference ran four separate Spark jobs *per column*
plus per-row string UDFs. Here all numeric screening stats (max, mean,
stddev, fractional-precision flag) come from fused tensor reductions on
device; geohash candidacy checks run over the column *dictionary* on
host (tiny), never over rows.
"""

from __future__ import annotations

from typing import List, Tuple

import torch

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_transformer import geo_utils as gu


def geo_to_latlong(x, option: int = 0):
    """Decode geohash string(s) → lat/long (reference geo_auto_detection.py:101
    via pygeohash; here the native bit-interleave decoder).
    option 0 → latitude, 1 → longitude."""
    single = isinstance(x, str)
    lst = [x] if single else list(x)
    if any(not gu.geohash_is_valid(str(g).lower()) for g in lst):
        raise ValueError("invalid geohash")
    ints, prec = gu.geohash_str_to_int([str(g).lower() for g in lst])
    t = torch.from_numpy(ints)
    lat, lon = gu.geohash_decode_int(t, prec)
    out = lat if option == 0 else lon
    vals = [float(v) for v in out]
    return vals[0] if single else vals


def latlong_to_geo(lat, long, precision: int = 9):
    """Encode lat/long → geohash string(s) (reference geo_auto_detection.py:143)."""
    single = not hasattr(lat, "__len__")
    la = torch.as_tensor([lat] if single else list(lat), dtype=torch.float64)
    lo = torch.as_tensor([long] if single else list(long), dtype=torch.float64)
    gh = gu.geohash_encode_int(la, lo, precision=precision)
    out = gu.geohash_int_to_str(gh, precision=precision)
    return out[0] if single else out


def _has_fraction(x: torch.Tensor) -> bool:
    v = x[~torch.isnan(x)]
    if v.numel() == 0:
        return False
    return bool(((v - v.trunc()).abs() > 1e-9).any())


def ll_gh_cols(df: AnovosFrame, max_records: int = 100_000) -> Tuple[List[str], List[str], List[str]]:
    """Detect (lat_cols, long_cols, gh_cols) with the reference's
    heuristics (geo_auto_detection.py:177-298): name match first; else a
    float column qualifies when it has decimal precision, |max| ≤ 90
    (latitude) or 90 < |max| ≤ 180 (longitude), stddev ≥ 1 and
    coefficient-of-variation < 1, and > 2 distinct values. String
    columns of length 5-11 whose values all geohash-decode qualify as
    geohash columns."""
    lat_cols, long_cols, gh_cols = [], [], []
    name_verdict = {}  # name-match detections (schema-only, rank-uniform)
    # data-driven verdicts that must be reconciled across ranks: the
    # reference's screens (max/std/mean/distinct, all-values-decode) are
    # Spark dataset-wide; per-shard verdicts can disagree (an empty
    # shard detects nothing) and a rank-dependent column list deadlocks
    # the collective stats that follow. Name-based detection is
    # schema-only and already rank-uniform.
    num_verdict = {}  # name -> "lat" | "long" (data-driven only)
    gh_state = {}  # name -> "valid" | "no" | "nodata"
    for name, dtype in df.dtypes:
        c = df.col(name)
        if c.kind == "numerical":
            lname = name.lower()
            if "latitude" in lname:
                name_verdict[name] = "lat"
                continue
            if "longitude" in lname:
                name_verdict[name] = "long"
                continue
            x = c.data.to(torch.float64)
            v = x[~torch.isnan(x)]
            if v.numel() < 3:
                continue
            mx = float(v.max())
            mn_abs_max = float(v.abs().max())
            sd = float(v.std())
            mean = float(v.mean())
            if not _has_fraction(x):
                continue
            if mn_abs_max > 180 or sd < 1 or mean == 0 or sd / abs(mean) >= 1:
                continue
            distinct = int(torch.unique(v).numel())
            if distinct <= 2:
                continue
            if mn_abs_max <= 90:
                num_verdict[name] = "lat"
            elif mn_abs_max <= 180:
                num_verdict[name] = "long"
        elif c.kind == "categorical":
            d = [s for s in (c.dictionary or []) if s][:max_records]
            if not d:
                gh_state[name] = "nodata"
                continue
            max_len = max(len(str(s)) for s in d)
            if (
                4 < max_len < 12
                and len(set(d)) > 2
                and all(gu.geohash_is_valid(str(s).lower()) for s in d)
            ):
                gh_state[name] = "valid"
            else:
                gh_state[name] = "no"

    from anovos_amd.core import dist as _dist

    if _dist.world_size() > 1:
        gathered = _dist.all_gather_object((num_verdict, gh_state))
        # numeric: a column is longitude if ANY rank saw |max| in
        # (90, 180] (the global |max| would too), else latitude if any
        # rank said lat. geohash: every rank holding data must validate.
        names = {n for nv, _ in gathered for n in nv}
        num_verdict = {}
        for n in names:
            vs = {nv.get(n) for nv, _ in gathered} - {None}
            num_verdict[n] = "long" if "long" in vs else "lat"
        gh_names = {n for _, gs in gathered for n in gs}
        gh_state = {}
        for n in gh_names:
            states = {gs.get(n, "nodata") for _, gs in gathered}
            if "no" in states or "valid" not in states:
                gh_state[n] = "no"
            else:
                gh_state[n] = "valid"

    # fold verdicts back in schema order (rank-uniform; preserves the
    # reference's lat↔long pairing order across both mechanisms)
    for name, _ in df.dtypes:
        v = name_verdict.get(name) or num_verdict.get(name)
        if v == "lat":
            lat_cols.append(name)
        elif v == "long":
            long_cols.append(name)
        elif gh_state.get(name) == "valid":
            gh_cols.append(name)
    if len(lat_cols) != len(long_cols):
        lat_cols, long_cols = [], []
    return lat_cols, long_cols, gh_cols


# ---------------------------------------------------------- parity helpers
# (reference geo_auto_detection.py:22-98)

def reg_lat_lon(option: str) -> str:
    """Regex for plausible latitude/longitude strings."""
    if option == "latitude":
        return r"^(\+|-|)?(?:90(?:(?:\.0{1,10})?)|(?:[0-9]|[1-8][0-9])(?:(?:\.[0-9]{1,})?))$"
    if option == "longitude":
        return r"^(\+|-)?(?:180(?:(?:\.0{1,10})?)|(?:[0-9]|[1-9][0-9]|1[0-7][0-9])(?:(?:\.[0-9]{1,10})?))$"
    raise ValueError("option must be latitude or longitude")


def conv_str_plus(col):
    """Prefix '+' onto non-negative values (regex normalization)."""
    if col is None:
        return None
    if col < 0:
        return col
    return "+" + str(col)


def precision_lev(col) -> int:
    """Number of significant decimal places (0 for integral/None)."""
    if col is None:
        return 0
    frac = format(float(col), ".8f").split(".")[1]
    if float(frac) > 0:
        return len(frac)
    return 0
