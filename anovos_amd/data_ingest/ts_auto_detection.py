"""Auto-detect & convert timestamp columns (reference parity:
``anovos/data_ingest/ts_auto_detection.py`` :51-761).

MI355X-native: candidate screening uses column metadata + dictionary
lengths (host, tiny); the actual parse runs once over the *dictionary*
of a string column (dateutil + strptime formats) or as an int64 scale
for 10/13-digit epoch columns, then applies on-GPU via LUT gather —
never a per-row UDF like the reference (:314-528).
"""

from __future__ import annotations

import datetime as _dt
import os
import re
from typing import List, Tuple

import pandas as pd
import torch

from anovos_amd.core.dtypes import NULL_CODE, NULL_TS
from anovos_amd.core.frame import AnovosFrame, Column
from anovos_amd.shared.utils import attributeType_segregation, ends_with

US_PER_SEC = 1_000_000

# candidate fixed widths the reference screens for (ts_auto_detection.py:554-620):
# 4=yyyy, 6=yyyymm, 8=yyyymmdd, 10=epoch-sec or yyyy-mm-dd, 13=epoch-ms
_CAND_LENGTHS = {4, 6, 8, 10, 13}

_EXPLICIT_FORMATS = [
    "%Y-%m-%d %H:%M:%S", "%Y-%m-%d %H:%M", "%Y-%m-%d",
    "%Y/%m/%d %H:%M:%S", "%Y/%m/%d", "%d-%m-%Y", "%d/%m/%Y",
    "%m/%d/%Y %H:%M:%S", "%m/%d/%Y", "%Y%m%d", "%Y%m", "%Y",
    "%d-%b-%Y", "%d %b %Y", "%b %d, %Y", "%Y-%m-%dT%H:%M:%S",
]

_RE_EPOCH10 = re.compile(r"^\d{10}$")
_RE_EPOCH13 = re.compile(r"^\d{13}$")
_RE_NONDATE = re.compile(r"^[a-zA-Z]+$|@|://")  # words, emails, urls never parse

# label -> regex map for candidate screening, same contract as the
# reference's REGEX_PATTERNS_PARSERS table (ts_auto_detection.py:151):
# a column qualifies as a timestamp candidate when its sampled values
# match one of these shapes before the strptime/dateutil parse attempts.
REGEX_PATTERNS_PARSERS = {
    "epoch_sec": r"^\d{10}$",
    "epoch_ms": r"^\d{13}$",
    "yyyy": r"^\d{4}$",
    "yyyymm": r"^\d{6}$",
    "yyyymmdd": r"^\d{8}$",
    "iso_date": r"^\d{4}-\d{2}-\d{2}$",
    "iso_datetime": r"^\d{4}-\d{2}-\d{2}[T ]\d{2}:\d{2}(:\d{2})?$",
    "slash_date": r"^\d{4}/\d{2}/\d{2}$|^\d{2}/\d{2}/\d{4}$",
    "dash_dmy": r"^\d{2}-\d{2}-\d{4}$",
    "month_name": r"^\d{1,2}[- ][A-Za-z]{3}[- ]\d{4}$|^[A-Za-z]{3} \d{1,2}, \d{4}$",
}


def _parse_one(s: str):
    """Parse a single string to epoch-us, or None."""
    if s is None or s == "" or _RE_NONDATE.search(str(s)):
        return None
    t = str(s).strip()
    if _RE_EPOCH10.match(t):
        return int(t) * US_PER_SEC
    if _RE_EPOCH13.match(t):
        return int(t) * 1000
    for f in _EXPLICIT_FORMATS:
        try:
            dt = _dt.datetime.strptime(t, f)
            return int((dt - _dt.datetime(1970, 1, 1)).total_seconds() * US_PER_SEC)
        except ValueError:
            continue
    try:
        from dateutil import parser as duparser

        dt = duparser.parse(t, fuzzy=False)
        if dt.tzinfo is not None:
            dt = dt.astimezone(_dt.timezone.utc).replace(tzinfo=None)
        return int((dt - _dt.datetime(1970, 1, 1)).total_seconds() * US_PER_SEC)
    except (ValueError, OverflowError, TypeError):
        return None


def regex_date_time_parser(ctx, idf: AnovosFrame, id_col: str = "", col: str = "",
                           tz: str = "local", val_unique_cat: int = 0,
                           trans_cat: str = "string", save_output=None,
                           output_mode: str = "replace", precision: str = "s"):
    """Reference ts_auto_detection.py:51 — convert one candidate column
    to a timestamp column. ≥80% of non-null distinct values must parse,
    else the column is returned untouched.

    Positional layout matches the reference: (ctx, idf, id_col, col, tz,
    val_unique_cat, trans_cat, save_output, output_mode). A two-arg call
    ``regex_date_time_parser(ctx, idf, "colname")`` is also accepted
    (id_col slot holding the column, reference col empty) for engine-
    internal use. ``save_output``: path to write the converted frame
    (reference :520-528)."""
    if not col:  # short form: (ctx, idf, col)
        col = id_col
    c = idf.col(col)
    dev = c.data.device
    if c.dtype in ("timestamp", "date"):
        return idf
    from anovos_amd.core import dist as _dist

    if c.kind == "numerical":
        x = c.data.to(torch.float64)
        null = torch.isnan(x)
        v = x[~null]
        # range/integrality decisions must be GLOBAL (the reference's
        # Spark min/max are dataset-wide): otherwise ranks with different
        # shards pick different branches and the resulting schemas (and
        # later collective counts) diverge. One batched all-reduce.
        mn = float(v.min()) if v.numel() else float("inf")
        mx = float(v.max()) if v.numel() else float("-inf")
        allint = 1.0 if (v.numel() == 0 or bool(((v - v.trunc()) == 0).all())) else 0.0
        if _dist.world_size() > 1:
            neg_mn, mx, neg_allint = _dist.all_reduce_scalars([-mn, mx, -allint], op="max")
            mn, allint = -neg_mn, -neg_allint
        if mx < mn:  # globally empty column
            return idf
        if 1e9 <= mn and mx < 1e10:  # 10-digit epoch seconds
            ts = torch.where(null, torch.zeros_like(x), x).to(torch.int64) * US_PER_SEC
        elif 1e12 <= mn and mx < 1e13:  # 13-digit epoch millis
            ts = torch.where(null, torch.zeros_like(x), x).to(torch.int64) * 1000
        elif 1000 <= mn and mx <= 9999 and allint >= 1.0:  # yyyy
            days = []
            yrs = v.to(torch.int64)
            epoch = _dt.datetime(1970, 1, 1)
            uniq = torch.unique(yrs)
            lut = {int(y): int((_dt.datetime(int(y), 1, 1) - epoch).total_seconds() * US_PER_SEC) for y in uniq if 1 <= int(y) <= 9999}
            tsv = torch.tensor([lut.get(int(y), NULL_TS) for y in yrs.cpu()], dtype=torch.int64)
            ts = torch.full_like(c.data.to(torch.int64), NULL_TS)
            ts[~null] = tsv.to(dev)
            ts = ts
        elif 19000101 <= mn and mx <= 29991231:  # yyyymmdd int
            ymd = torch.where(null, torch.zeros_like(x), x).to(torch.int64)
            y = torch.div(ymd, 10000, rounding_mode="floor")
            m = torch.div(ymd % 10000, 100, rounding_mode="floor")
            d = ymd % 100
            from anovos_amd.data_transformer.datetime import _civil_to_days, US_PER_DAY

            ok = (m >= 1) & (m <= 12) & (d >= 1) & (d <= 31)
            ts = _civil_to_days(y, m.clamp(1, 12), d.clamp(1, 31)) * US_PER_DAY
            null = null | ~ok
        else:
            return idf
        ts = torch.where(null | (ts == NULL_TS), torch.full_like(ts, NULL_TS), ts)
        newc = Column(col + "_ts", "timestamp", ts)
    elif c.kind == "categorical":
        d = c.dictionary or []
        parsed = [_parse_one(s) for s in d]
        n_nonnull = sum(1 for s in d if s)
        n_ok = sum(1 for p in parsed if p is not None)
        # the ≥80% acceptance rule is over the GLOBAL distinct values
        # (dictionaries are per-rank): sum counts across ranks so every
        # rank reaches the same convert/skip verdict.
        if _dist.world_size() > 1:
            n_nonnull, n_ok = _dist.all_reduce_scalars([float(n_nonnull), float(n_ok)], op="sum")
        if n_nonnull == 0 or n_ok / max(n_nonnull, 1) < 0.8:
            return idf
        lut = torch.tensor([p if p is not None else NULL_TS for p in parsed] + [NULL_TS],
                           dtype=torch.int64, device=dev)
        codes = c.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(d)), codes)
        newc = Column(col + "_ts", "timestamp", lut[codes])
    else:
        return idf
    if output_mode == "replace":
        odf = idf.with_column(col, Column(col, "timestamp", newc.data))
    else:
        odf = idf.with_column(col + "_ts", newc)
    if save_output:
        from anovos_amd.core import io as _io

        _io.write_dataset(odf, str(save_output), "parquet", {"mode": "overwrite"})
    return odf


def ts_loop_cols_pre(idf: AnovosFrame, id_col: str) -> Tuple[List[str], List[str], List[int]]:
    """Reference ts_auto_detection.py:554 — candidate classification by
    dtype + fixed string/number width ∈ {4,6,8,10,13}.

    The width/distinct screening stats are merged across ranks (len-set
    union, max width, summed distinct counts, OR'd non-integrality): in
    the reference these are Spark dataset-wide aggregates, and every
    rank must classify identically or later per-ts-column collective
    loops deadlock (empty or skewed shards would otherwise disagree)."""
    from anovos_amd.core import dist as _dist

    # pass 1 — local per-column screening stats (schema order is
    # identical on every rank)
    meta = []  # (name, dtype, kind, col_len, lens, distinct, nonint)
    for name, dtype in idf.dtypes:
        c = idf.col(name)
        if c.kind == "categorical":
            d = [s for s in (c.dictionary or []) if s]
            col_len = max((len(str(s)) for s in d), default=0)
            lens = {len(str(s)) for s in d}
            nonint = False
        elif c.kind == "numerical":
            x = c.data
            v = x[~torch.isnan(x)] if x.is_floating_point() else x
            if v.numel() and bool(((v - v.trunc()) == 0).all()):
                iv = v.to(torch.int64)
                strs_len = torch.where(iv == 0, torch.ones_like(iv), torch.log10(iv.abs().clamp(min=1).to(torch.float64)).to(torch.int64) + 1)
                col_len = int(strs_len.max())
                lens = set(strs_len.unique().cpu().numpy().tolist())
                nonint = False
            else:
                col_len, lens, nonint = 0, set(), bool(v.numel())
        else:
            meta.append((name, dtype, "other", 0, set(), 0, False))
            continue
        nonnull_distinct = int(torch.unique(c.data[~c.null_mask()]).numel())
        meta.append((name, dtype, c.kind, col_len, lens, nonnull_distinct, nonint))

    # pass 2 — merge across ranks (one object gather; ingest-time cold path)
    if _dist.world_size() > 1:
        gathered = _dist.all_gather_object([(m[3], sorted(m[4]), m[5], m[6]) for m in meta])
        merged = []
        for i, m in enumerate(meta):
            col_len = max(g[i][0] for g in gathered)
            lens = set()
            for g in gathered:
                lens.update(g[i][1])
            distinct = sum(g[i][2] for g in gathered)
            nonint = any(g[i][3] for g in gathered)
            if nonint:  # fractional values on some rank → not a fixed-width int candidate
                col_len, lens = 0, set()
            merged.append((m[0], m[1], m[2], col_len, lens, distinct, nonint))
        meta = merged

    # pass 3 — classify (pure function of merged stats: rank-uniform)
    lc1, lc2, lc3 = [], [], []
    for name, dtype, kind, col_len, lens, distinct, _nonint in meta:
        if kind == "other":
            lc1.append(name)
            lc2.append("dt" if dtype in ("timestamp", "date") else "NA")
            lc3.append(0)
        elif distinct == 0:
            lc1.append(name); lc2.append("NA"); lc3.append(col_len)
        elif name != id_col and len(lens) == 1 and col_len in _CAND_LENGTHS:
            tag = "string_c" if kind == "categorical" else ("bigint_c" if dtype in ("bigint", "long", "double") else "int_c")
            lc1.append(name); lc2.append(tag); lc3.append(col_len)
        elif name != id_col and kind == "categorical":
            lc1.append(name); lc2.append("string"); lc3.append(col_len)
        else:
            lc1.append(name); lc2.append("NA"); lc3.append(col_len)
    return lc1, lc2, lc3


def ts_preprocess(ctx, idf: AnovosFrame, id_col: str, output_path: str,
                  tz_offset: str = "local", run_type: "str" = "local", mlflow_config=None, auth_key="NA"):
    """Reference ts_auto_detection.py:622 — driver: convert candidate
    columns, write ``ts_cols_stats.csv``, return (odf, ts_cols)."""
    from anovos_amd.core import dist as _dist

    local_path = output_path if run_type == "local" else "report_stats"
    if _dist.rank() == 0:
        os.makedirs(local_path, exist_ok=True)
    _dist.barrier()
    lc1, lc2, lc3 = ts_loop_cols_pre(idf, id_col)
    ts_loop_cols = [lc1[i] for i, k in enumerate(lc2) if k in ("string", "string_c", "int_c", "bigint_c", "long_c")]
    pre_exist_ts_cols = [lc1[i] for i, k in enumerate(lc2) if k == "dt"]
    odf = idf
    for i in ts_loop_cols:
        try:
            odf = regex_date_time_parser(ctx, odf, i, output_mode="replace")
        except Exception:
            continue  # reference swallows per-column parse failures (:694-709)
    # reconcile: conversion decisions are already global (see
    # regex_date_time_parser / ts_loop_cols_pre), but a swallowed
    # per-column exception on one rank could still leave the schemas
    # diverged — and a rank-dependent ts_cols set deadlocks every later
    # per-column collective loop (ts_analyzer). Keep the intersection,
    # reverting any local-only conversion to the original column.
    local_post = [n for n, d in odf.dtypes if d in ("timestamp", "date")]
    if _dist.world_size() > 1:
        sets = _dist.all_gather_object(sorted(local_post))
        agreed = set(sets[0])
        for s in sets[1:]:
            agreed &= set(s)
        for n in local_post:
            if n not in agreed:
                odf = odf.with_column(n, idf.col(n))
    ts_cols_post = [n for n, d in odf.dtypes if d in ("timestamp", "date")]
    num_cols, cat_cols, other_cols = attributeType_segregation(odf)
    num_cols = [x for x in num_cols if x not in [id_col] + ts_cols_post]
    cat_cols = [x for x in cat_cols if x not in [id_col] + ts_cols_post]
    auto_detected = sorted(set(ts_cols_post) - set(pre_exist_ts_cols))
    stats = pd.DataFrame(
        {
            "attribute": ts_cols_post,
            "source": ["pre_existing" if c in pre_exist_ts_cols else "auto_detected" for c in ts_cols_post],
        }
    )
    if _dist.rank() == 0:
        stats.to_csv(ends_with(local_path) + "ts_cols_stats.csv", index=False)
    _dist.barrier()
    return odf, ts_cols_post, num_cols, cat_cols
