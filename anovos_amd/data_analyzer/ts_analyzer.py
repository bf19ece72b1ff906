"""Time-series inspection over detected timestamp columns (reference
parity: ``anovos/data_analyzer/ts_analyzer.py`` :52-551, same on-disk
contract: ``stats_<col>_{1,2}.csv`` and ``<ts>_<col>_<type>.csv`` under
the output path).

MI355X-native: unit extraction and lag-diff stats are int64 tensor ops
(data_transformer/datetime.py); per-bucket aggregations run via
scatter_reduce on the GPU; only the tiny result tables land on host.
"""

from __future__ import annotations

import os
from typing import Optional

import numpy as np
import pandas as pd
import torch

from anovos_amd.core.dtypes import NULL_CODE, NULL_TS
from anovos_amd.core.frame import AnovosFrame, Column
from anovos_amd.data_transformer.datetime import (
    US_PER_DAY,
    _decompose,
    _floor_day,
    timeUnits_extraction,
)
from anovos_amd.shared.utils import attributeType_segregation, ends_with

DAYPARTS = ["early_hours", "work_hours", "late_hours", "commuting_hours", "other_hours", "Missing_NA"]


def daypart_cat(column):
    """Reference ts_analyzer.py:52 — scalar hour → daypart label
    (reference arg name: column)."""
    hour = column
    if hour is None:
        return "Missing_NA"
    if 4 <= hour < 7:
        return "early_hours"
    if 10 <= hour < 17:
        return "work_hours"
    if hour >= 23 or hour < 4:
        return "late_hours"
    if 7 <= hour < 10 or 17 <= hour < 20:
        return "commuting_hours"
    return "other_hours"


def _daypart_codes(hh: torch.Tensor, null: torch.Tensor) -> torch.Tensor:
    """Vectorized daypart bucketing (codes into DAYPARTS)."""
    out = torch.full_like(hh, 4)  # other_hours
    out = torch.where((hh >= 4) & (hh < 7), torch.zeros_like(out), out)
    out = torch.where((hh >= 10) & (hh < 17), torch.ones_like(out), out)
    out = torch.where((hh >= 23) | (hh < 4), torch.full_like(out, 2), out)
    out = torch.where(((hh >= 7) & (hh < 10)) | ((hh >= 17) & (hh < 20)), torch.full_like(out, 3), out)
    out = torch.where(null, torch.full_like(out, 5), out)
    return out


def ts_processed_feats(idf: AnovosFrame, col: str, id_col: str, tz: str = "local",
                       cnt_row: Optional[int] = None, cnt_unique_id: Optional[int] = None) -> AnovosFrame:
    """Reference ts_analyzer.py:87 — append unit columns + yyyymmdd_col,
    daypart_cat, week_cat, dow."""
    odf = timeUnits_extraction(idf, [col], "all", output_mode="append")
    ts = idf.col(col).data
    null = ts == NULL_TS
    day = _floor_day(ts)
    odf = odf.with_column("yyyymmdd_col", Column("yyyymmdd_col", "date",
                                                 torch.where(null, torch.full_like(ts, NULL_TS), day * US_PER_DAY)))
    hh = odf.col(col + "_hour").data
    hh_null = torch.isnan(hh)
    dp = _daypart_codes(torch.nan_to_num(hh).to(torch.int64), hh_null | null)
    odf = odf.with_column("daypart_cat", Column("daypart_cat", "string", dp.to(torch.int32), list(DAYPARTS)))
    dow = odf.col(col + "_dayofweek").data
    wk = torch.where(torch.nan_to_num(dow) > 5, torch.ones_like(dow), torch.zeros_like(dow))
    wk_codes = wk.to(torch.int32)
    wk_codes = torch.where(null, torch.full_like(wk_codes, NULL_CODE), wk_codes)
    odf = odf.with_column("week_cat", Column("week_cat", "string", wk_codes, ["weekday", "weekend"]))
    odf = odf.rename({col + "_dayofweek": "dow"})
    return odf


def ts_eligiblity_check(ctx, idf: AnovosFrame, id_col: str, opt: int = 1, tz_offset: str = "local") -> pd.DataFrame:
    """Reference ts_analyzer.py:160 — opt=1: percentile stats of
    dates-per-id and ids-per-date pair counts (two-row table). opt=2:
    one-row summary [count_unique_dates, min_date, max_date, modal_date,
    date_diff, missing_date, mean, variance, stdev, cov] where the last
    four are lag-1 day-diff statistics over the distinct dates (rounded
    to 3, like the reference)."""
    from anovos_amd.core import dist as _dist

    day = idf.col("yyyymmdd_col").data
    valid = day != NULL_TS
    days = torch.unique(day[valid])
    if _dist.is_dist():
        days = torch.unique(torch.cat(_dist.all_gather_tensor(days)))
    if opt == 1:
        # distribution of dates-per-id and ids-per-date (reference p1∪p2)
        idc = idf.col(id_col)
        id_codes = idc.data.to(torch.long) if idc.kind == "categorical" else torch.unique(idc.data, return_inverse=True)[1]
        ok = valid & ~idc.null_mask()
        pair = torch.stack([id_codes[ok].to(torch.float64), day[ok].to(torch.float64)], dim=1)
        uniq_pair = torch.unique(pair, dim=0)
        if _dist.is_dist():
            uniq_pair = torch.unique(torch.cat(_dist.all_gather_tensor(uniq_pair)), dim=0)
        rows = []
        for key_idx, name in ((0, "id_date_pair"), (1, "date_id_pair")):
            keys = uniq_pair[:, key_idx]
            _, counts = torch.unique(keys, return_counts=True)
            c = counts.to(torch.float64)
            qs = torch.quantile(c, torch.tensor([0.01, 0.05, 0.1, 0.25, 0.5, 0.75, 0.9, 0.95, 0.99],
                                                dtype=torch.float64, device=c.device))
            rows.append([name, float(c.min()), *[float(q) for q in qs], float(c.max())])
        return pd.DataFrame(rows, columns=["attribute", "min", "1%", "5%", "10%", "25%", "50%", "75%", "90%", "95%", "99%", "max"])

    # opt == 2: one-row date summary + lag-1 diff stats (reference :230-257)
    if days.numel() > 1:
        diffs = (days[1:] - days[:-1]).to(torch.float64) / US_PER_DAY
        mean = round(float(diffs.mean()), 3)
        var = round(float(diffs.var(unbiased=True)) if diffs.numel() > 1 else 0.0, 3)
        sd = round(var ** 0.5, 3)
        cov = round(sd / mean, 3) if mean else float("nan")
    else:
        mean = var = sd = cov = float("nan")
    # global per-date row counts for the modal date
    uniq_d, cnt_d = torch.unique(day[valid], return_counts=True)
    if _dist.is_dist():
        gv = torch.cat(_dist.all_gather_tensor(uniq_d))
        gc = torch.cat(_dist.all_gather_tensor(cnt_d.to(torch.int64)))
        uniq_d, inv = torch.unique(gv, return_inverse=True)
        cnt_d = torch.zeros(uniq_d.numel(), dtype=torch.int64, device=gv.device)
        cnt_d.index_add_(0, inv, gc)
    missing = int(_dist.all_reduce_scalar(int((~valid).sum())))

    def _date(us):
        return pd.Timestamp(int(us), unit="us").date()

    if uniq_d.numel():
        best_cnt = int(cnt_d.max())
        cand = uniq_d[cnt_d == best_cnt]
        modal = str(_date(int(cand.min()))) + " [" + str(best_cnt) + "]"
        min_d, max_d = _date(int(days.min())), _date(int(days.max()))
        date_diff = (max_d - min_d).days
    else:
        modal, min_d, max_d, date_diff = None, None, None, 0
    return pd.DataFrame({
        "count_unique_dates": [int(days.numel())],
        "min_date": [min_d],
        "max_date": [max_d],
        "modal_date": [modal],
        "date_diff": [date_diff],
        "missing_date": [missing],
        "mean": [mean],
        "variance": [var],
        "stdev": [sd],
        "cov": [cov],
    })


def ts_viz_data(idf: AnovosFrame, x_col: str, y_col: str, id_col: str = "",
                tz_offset="local", output_mode="append", output_type="daily",
                n_cat: int = 10) -> pd.DataFrame:
    """Reference ts_analyzer.py:259 — per (bucket × column) aggregates:
    counts for categorical y, min/max/mean/median for numeric y.
    Multi-rank: keys are unified and count/sum/min/max all-reduced; the
    per-bucket median is computed over the local shard (a full global
    median would need per-bucket value exchange — report-viz tolerance)."""
    from anovos_amd.core import dist as _dist

    key_map = {"daily": "yyyymmdd_col", "hourly": "daypart_cat", "weekly": "dow"}
    k_col = key_map[output_type]
    key = idf.col(k_col)
    yc = idf.col(y_col)
    if yc.kind == "categorical":  # schema-based → rank-uniform call
        from anovos_amd.ops.groupby import align_dictionaries

        align_dictionaries(idf, [y_col])
    if key.kind == "categorical":
        knull = key.data == NULL_CODE
        kcodes = key.data.to(torch.long)
        klabels = key.dictionary
    elif key.dtype in ("timestamp", "date"):
        knull = key.data == NULL_TS
        uniq, kcodes = torch.unique(key.data, return_inverse=True)
        klabels = [str(pd.Timestamp(int(u), unit="us").date()) if int(u) != NULL_TS else None for u in uniq]
    else:
        knull = torch.isnan(key.data)
        uniq, kcodes = torch.unique(torch.nan_to_num(key.data), return_inverse=True)
        klabels = [float(u) for u in uniq]
    if _dist.is_dist() and key.kind != "categorical":
        # unify bucket keys across ranks (dates/dows differ per shard)
        gt = torch.unique(torch.cat(_dist.all_gather_tensor(uniq)))
        guniq = gt.cpu().numpy()
        pos = torch.searchsorted(gt, uniq)
        kcodes = pos[kcodes]
        if key.dtype in ("timestamp", "date"):
            klabels = [str(pd.Timestamp(int(u), unit="us").date()) if int(u) != NULL_TS else None for u in guniq]
        else:
            klabels = [float(u) for u in guniq]
    G = len(klabels)
    if G == 0:  # globally 0-row frame; rank-uniform (keys were unified)
        if yc.kind == "categorical":
            return pd.DataFrame(columns=[y_col, k_col, "count"])
        return pd.DataFrame(columns=[k_col, "min", "max", "mean", "median"])
    dev = kcodes.device
    if yc.kind == "categorical":
        # top-n_cat categories by count, then count per (bucket, cat)
        ynull = yc.data == NULL_CODE
        yv = yc.data.to(torch.long)
        m = ~knull & ~ynull
        nd = len(yc.dictionary or [])
        if nd == 0:  # rank-uniform: dictionaries are aligned above
            return pd.DataFrame(columns=[y_col, k_col, "count"])
        # the top-n_cat choice (and the zero-count skip) must come from
        # the GLOBAL counts — a local choice gives each rank a different
        # reduce sequence and deadlocks. One [n_cat, G] fused count +
        # ONE all-reduce instead of a reduce per category.
        cnt_y = torch.bincount(yv[m], minlength=nd).to(torch.float64)
        _dist.all_reduce_(cnt_y, "sum")
        top = torch.argsort(cnt_y, descending=True, stable=True)[:n_cat]
        tpos = torch.full((nd,), -1, dtype=torch.long, device=dev)
        tpos[top] = torch.arange(top.numel(), device=dev)
        sel = m & (tpos[yv] >= 0)
        comb = tpos[yv[sel]] * G + kcodes[sel]
        per = torch.zeros(top.numel() * G, dtype=torch.float64, device=dev)
        per.scatter_reduce_(0, comb, torch.ones(comb.numel(), dtype=torch.float64, device=dev), reduce="sum")
        _dist.all_reduce_(per, "sum")
        per = per.view(top.numel(), G).cpu()
        rows = []
        for ti, t in enumerate(top.cpu().numpy()):
            if float(cnt_y[t]) == 0:
                continue
            for g in range(G):
                if float(per[ti, g]) > 0:
                    rows.append([yc.dictionary[int(t)], klabels[g], float(per[ti, g])])
        return pd.DataFrame(rows, columns=[y_col, k_col, "count"])
    x = yc.data.to(torch.float64)
    ynull = torch.isnan(x)
    m = ~knull & ~ynull
    xv, kv = x[m], kcodes[m]
    cnt = torch.zeros(G, dtype=torch.float64, device=dev).scatter_reduce(0, kv, torch.ones_like(xv), reduce="sum")
    s = torch.zeros(G, dtype=torch.float64, device=dev).scatter_reduce(0, kv, xv, reduce="sum")
    mn = torch.full((G,), float("inf"), dtype=torch.float64, device=dev).scatter_reduce(0, kv, xv, reduce="amin")
    mx = torch.full((G,), float("-inf"), dtype=torch.float64, device=dev).scatter_reduce(0, kv, xv, reduce="amax")
    _dist.all_reduce_(cnt, "sum")
    _dist.all_reduce_(s, "sum")
    _dist.all_reduce_(mn, "min")
    _dist.all_reduce_(mx, "max")
    # median: sort values, stable-sort groups, middle of each segment
    byval = torch.argsort(xv)
    bygrp = torch.argsort(kv[byval], stable=True)
    o = byval[bygrp]
    gs, xs = kv[o].contiguous(), xv[o]
    starts = torch.searchsorted(gs, torch.arange(G, device=dev))
    # interpolated median (pandas convention, like the reference): mean
    # of the two middle elements for even group sizes
    cl = cnt.to(torch.long)
    cap = max(int(xs.shape[0]) - 1, 0)
    lo = (starts + ((cl - 1) // 2).clamp(min=0)).clamp(max=cap)
    hi = (starts + (cl // 2).clamp(min=0)).clamp(max=cap)
    med = (xs[lo] + xs[hi]) / 2 if xs.numel() else torch.zeros(G, dtype=torch.float64)
    pdf = pd.DataFrame(
        {
            k_col: klabels,
            "min": mn.cpu().numpy(),
            "max": mx.cpu().numpy(),
            "mean": (s / cnt.clamp(min=1)).cpu().numpy(),
            "median": med.cpu().numpy() if xs.numel() else np.zeros(G),
        }
    )
    return pdf[cnt.cpu().numpy() > 0].reset_index(drop=True)


def ts_analyzer(ctx, idf: AnovosFrame, id_col: str, max_days: int, output_path: str,
                output_type: str = "daily", tz_offset: str = "local", run_type: str = "local", auth_key="NA"):
    """Reference ts_analyzer.py:408 — driver loop: for each ts column
    write eligibility stats (stats_<col>_{1,2}.csv) and per-attribute viz
    aggregates (<ts>_<attr>_<type>.csv)."""
    from anovos_amd.core import dist as _dist

    local_path = output_path if run_type == "local" else "report_stats"
    if _dist.rank() == 0:
        os.makedirs(local_path, exist_ok=True)
    _dist.barrier()
    write = _dist.rank() == 0
    num_cols, cat_cols, other = attributeType_segregation(idf)
    num_cols = [x for x in num_cols if x != id_col]
    cat_cols = [x for x in cat_cols if x != id_col]
    ts_cols = [n for n, d in idf.dtypes if d in ("timestamp", "date")]
    cnt_row = idf.count()
    for i in ts_cols:
        pdf_feats = ts_processed_feats(idf, i, id_col, tz_offset, cnt_row, None)
        f1 = ts_eligiblity_check(ctx, pdf_feats, id_col, opt=1)
        f2 = ts_eligiblity_check(ctx, pdf_feats, id_col, opt=2)
        if write:
            f1.to_csv(ends_with(local_path) + "stats_" + str(i) + "_1.csv", index=False)
            f2.to_csv(ends_with(local_path) + "stats_" + str(i) + "_2.csv", index=False)
        for cols in (num_cols, cat_cols):
            for l in cols:
                try:
                    f = ts_viz_data(pdf_feats, i, l, output_type=output_type, tz_offset=tz_offset)
                    if write:
                        f.to_csv(ends_with(local_path) + i + "_" + l + "_" + output_type + ".csv", index=False)
                except Exception:
                    continue
    return ts_cols
