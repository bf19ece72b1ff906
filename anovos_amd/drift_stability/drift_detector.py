"""Covariate-drift metrics — parity with reference
drift_stability/drift_detector.py (371 LoC; SURVEY.md §2.4, kernel K7).

Flow mirrors the reference (drift_detector.py:187-356): optional
sampling, bin SOURCE saving a cutoff model, bin TARGET with the saved
model, then per column p = source bin frequencies, q = target; PSI / HD /
JSD / KS over the (p,q) vectors with 0.0001 fill. The GPU path builds
ALL column histograms in one fused launch + one all-reduce; the metric
arithmetic runs on the tiny (bin_size+1)-vectors.
"""

from __future__ import annotations

import os

import numpy as np
import pandas as pd
import torch

from anovos_amd.core import dist
from anovos_amd.core.dtypes import NULL_CODE
from anovos_amd.data_ingest.data_sampling import data_sample
from anovos_amd.data_transformer.transformers import attribute_binning
from anovos_amd.drift_stability.validations import check_distance_method, check_list_of_columns
from anovos_amd.shared.utils import attributeType_segregation
from anovos_amd.shared.tracing import traced


@check_distance_method
@check_list_of_columns(target_idx=1, target="idf_target")
@traced
def statistics(
    ctx,
    idf_target,
    idf_source,
    *,
    list_of_cols="all",
    drop_cols=None,
    method_type="PSI",
    bin_method="equal_range",
    bin_size=10,
    threshold=0.1,
    use_sampling=True,
    sample_method="random",
    strata_cols="all",
    stratified_type="population",
    sample_size=100000,
    sample_seed=42,
    persist=True,
    persist_option=None,
    pre_existing_source=False,
    source_save=True,
    source_path="NA",
    model_directory="drift_statistics",
    print_impact=False,
):
    """Returns [attribute, <metrics...>, flagged] — reference
    drift_detector.py:18-371."""
    drop_cols = drop_cols or []
    # drift requires the column in BOTH frames; columns only present in
    # the target (e.g. joined-in side columns) are skipped with a warning
    absent = [c for c in list_of_cols if c not in idf_source.columns]
    if absent:
        import warnings

        warnings.warn("Columns not present in the source dataset are excluded from drift: " + ", ".join(absent))
        list_of_cols = [c for c in list_of_cols if c not in absent]
    num_cols = attributeType_segregation(idf_target.select(list_of_cols))[0]

    count_target = idf_target.count()
    count_source = idf_source.count()
    if use_sampling:
        if count_target > sample_size:
            idf_target = data_sample(
                idf_target,
                strata_cols=strata_cols,
                fraction=sample_size / count_target,
                method_type=sample_method,
                stratified_type=stratified_type,
                seed_value=sample_seed,
            )
            count_target = idf_target.count()
        if count_source > sample_size:
            idf_source = data_sample(
                idf_source,
                strata_cols=strata_cols,
                fraction=sample_size / count_source,
                method_type=sample_method,
                stratified_type=stratified_type,
                seed_value=sample_seed,
            )
            count_source = idf_source.count()

    if source_path == "NA":
        source_path = "intermediate_data"
    model_path = source_path + "/" + model_directory

    if not pre_existing_source:
        source_bin = attribute_binning(
            ctx, idf_source, list_of_cols=num_cols, method_type=bin_method,
            bin_size=bin_size, pre_existing_model=False, model_path=model_path,
        )
    else:
        source_bin = None
    target_bin = attribute_binning(
        ctx, idf_target, list_of_cols=num_cols, method_type=bin_method,
        bin_size=bin_size, pre_existing_model=True, model_path=model_path,
    )

    # batched frequency pass: one fused histogram launch covers every
    # numeric (binned) column of each frame
    q_freqs = batched_bin_frequencies(target_bin, list_of_cols, count_target, max_bin=bin_size)
    if not pre_existing_source:
        p_freqs = batched_bin_frequencies(source_bin, list_of_cols, count_source, max_bin=bin_size)
    rows = []
    for i in list_of_cols:
        if pre_existing_source:
            x = pd.read_csv(os.path.join(model_path, "frequency_counts", i, "part-00000.csv"))
            p_keys = [str(k) for k in x[i].tolist()]
            p_vals = x["p"].tolist()
        else:
            p_keys, p_vals = p_freqs[i]
            if source_save and dist.rank() == 0:
                d = os.path.join(model_path, "frequency_counts", i)
                os.makedirs(d, exist_ok=True)
                pd.DataFrame({i: p_keys, "p": p_vals}).to_csv(os.path.join(d, "part-00000.csv"), index=False)
        q_keys, q_vals = q_freqs[i]

        # full-outer join on bin key, fill 0.0001, order by key
        pmap = dict(zip(p_keys, p_vals))
        qmap = dict(zip(q_keys, q_vals))
        keys = sorted(set(pmap) | set(qmap), key=_key_order)
        p = np.array([max(pmap.get(k, 0.0001), 0.0001) if pmap.get(k, 0.0001) != 0 else 0.0001 for k in keys])
        q = np.array([max(qmap.get(k, 0.0001), 0.0001) if qmap.get(k, 0.0001) != 0 else 0.0001 for k in keys])
        p = np.where(p == 0, 0.0001, p)
        q = np.where(q == 0, 0.0001, q)

        row = {"attribute": i}
        if "PSI" in method_type:
            row["PSI"] = float(np.sum((p - q) * np.log(p / q)))
        if "HD" in method_type:
            row["HD"] = float(np.sqrt(np.sum((np.sqrt(p) - np.sqrt(q)) ** 2) / 2))
        if "JSD" in method_type:
            m = (p + q) / 2
            row["JSD"] = float((np.sum(p * np.log(p / m)) + np.sum(q * np.log(q / m))) / 2)
        if "KS" in method_type:
            row["KS"] = float(np.max(np.abs(np.cumsum(p) - np.cumsum(q))))
        rows.append(row)

    odf = pd.DataFrame(rows)
    metric_cols = [c for c in odf.columns if c != "attribute"]
    odf[metric_cols] = odf[metric_cols].round(4)
    odf["flagged"] = (odf[metric_cols] > threshold).any(axis=1).astype(int)
    if print_impact:
        print("All Attributes:")
        print(odf.to_string(index=False))
        print("Attributes meeting Data Drift threshold:")
        print(odf[odf["flagged"] == 1].to_string(index=False))
    return odf


def _key_order(k):
    try:
        return (0, float(k), "")
    except (TypeError, ValueError):
        return (1, 0.0, str(k))


def batched_bin_frequencies(binned_idf, cols, total: int, max_bin: int = 0):
    """Per-bin frequencies for many columns at once: numeric (binned)
    columns go through ONE fused histogram kernel + one all-reduce;
    categorical columns through the fused dictionary bincount. Pass
    ``max_bin`` (= the binning bin_size) to skip the moments pass — the
    per-column null count then falls out of the histogram row sum."""
    from anovos_amd.ops import histogram as hist_ops
    from anovos_amd.ops import stats as stats_ops

    out = {}
    num_cols = [c for c in cols if binned_idf.col(c).kind == "numerical"]
    cat_cols = [c for c in cols if c not in num_cols]
    if num_cols:
        if max_bin:
            M = int(max_bin)
            ns = None
        else:
            moments = stats_ops.frame_moments(binned_idf, num_cols)
            M = 1
            for c in num_cols:
                if moments[c].max == moments[c].max:
                    M = max(M, int(moments[c].max))
            ns = {c: moments[c].n for c in num_cols}
        tensors = [binned_idf.col(c).data for c in num_cols]
        lo = torch.full((len(num_cols),), 1.0, dtype=torch.float64)
        hi = torch.full((len(num_cols),), float(M + 1), dtype=torch.float64)
        hist = hist_ops.global_histograms(tensors, lo, hi, M).cpu().numpy()
        for i, c in enumerate(num_cols):
            keys, vals = [], []
            n_valid = float(hist[i].sum()) if ns is None else ns[c]
            nnull = int(total - n_valid)
            if nnull:
                keys.append("-1")
                vals.append(nnull / total)
            h = hist[i].tolist()
            for b, hb in enumerate(h):
                if hb > 0:
                    keys.append(str(b + 1))
                    vals.append(hb / total)
            out[c] = (keys, vals)
    for c in cat_cols:
        out[c] = _bin_frequencies(binned_idf, c, total)
    return out


def _bin_frequencies(binned_idf, col: str, total: int):
    """Global per-bin frequency of a (binned or categorical) column; null
    group keyed -1 (the reference's fillna(-1))."""
    c = binned_idf.col(col)
    if c.kind == "categorical":
        from anovos_amd.ops.groupby import cat_value_counts

        counts = cat_value_counts(binned_idf, [col])[col]
        nnull = int(dist.all_reduce_scalar(int((c.data == NULL_CODE).sum())))
        keys, vals = [], []
        for i, s in enumerate(c.dictionary or []):
            if counts[i] > 0:
                keys.append(str(s))
                vals.append(float(counts[i]) / total)
        if nnull:
            keys.append("-1")
            vals.append(nnull / total)
        return keys, vals
    t = c.data
    vals_t = torch.nan_to_num(t, nan=-1.0)
    iv = vals_t.to(torch.long)
    mx = int(dist.all_reduce_scalar(int(iv.max().item()) if iv.numel() else 0, "max"))
    cnt = torch.bincount((iv + 1).clamp(min=0), minlength=mx + 2).to(torch.float64)
    dist.all_reduce_(cnt, "sum")
    keys, vals = [], []
    for b, cb in enumerate(cnt.cpu().numpy().tolist()):
        if cb > 0:
            keys.append(str(b - 1))
            vals.append(cb / total)
    return keys, vals
