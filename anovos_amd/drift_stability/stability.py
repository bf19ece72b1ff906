"""Stability index across dataset snapshots — parity with reference
drift_stability/stability.py (589 LoC).

stability_index_computation: per attribute per snapshot compute
mean/stddev/kurtosis(+3) (one fused moments launch covering every
snapshot's columns), union with persisted historical metrics, CV of each
metric across snapshots, compute_si weighted score.
feature_stability_estimation: first/second-order Taylor propagation of
(mean, stddev) through a sympy formula (reference stability.py:421-439).
"""

from __future__ import annotations

import os

import numpy as np
import pandas as pd

from anovos_amd.core import dist
from anovos_amd.drift_stability.validations import (
    check_metric_weightages,
    check_threshold,
    compute_si,
)
from anovos_amd.ops import stats as stats_ops
from anovos_amd.shared.utils import attributeType_segregation


def stability_index_computation(
    ctx,
    *idfs,
    list_of_cols="all",
    drop_cols=[],
    metric_weightages={"mean": 0.5, "stddev": 0.3, "kurtosis": 0.2},
    binary_cols=[],
    existing_metric_path="",
    appended_metric_path="",
    persist=True,
    persist_option=None,
    threshold=1,
    print_impact=False,
):
    """Returns [attribute, type, mean_stddev, mean_cv, stddev_cv,
    kurtosis_cv, mean_si, stddev_si, kurtosis_si, stability_index,
    flagged] — reference stability.py:15-333."""
    num_cols_all = attributeType_segregation(idfs[0])[0]
    if list_of_cols == "all":
        list_of_cols = num_cols_all
    if isinstance(list_of_cols, str):
        list_of_cols = [x.strip() for x in list_of_cols.split("|")]
    if isinstance(drop_cols, str):
        drop_cols = [x.strip() for x in drop_cols.split("|")]
    if isinstance(binary_cols, str):
        binary_cols = [x.strip() for x in binary_cols.split("|")]
    list_of_cols = [e for e in dict.fromkeys(list_of_cols) if e not in drop_cols]
    if any(x not in num_cols_all for x in list_of_cols) or len(list_of_cols) == 0:
        raise TypeError("Invalid input for Column(s)")
    if any(x not in list_of_cols for x in binary_cols):
        raise TypeError("Invalid input for Binary Column(s)")
    check_metric_weightages(metric_weightages)
    check_threshold(threshold)

    if existing_metric_path:
        existing_metric_df = pd.read_csv(existing_metric_path) if os.path.isfile(existing_metric_path) else _read_csv_dir(existing_metric_path)
        dfs_count = int(existing_metric_df["idx"].max()) + 1
    else:
        existing_metric_df = None
        dfs_count = 1

    # one fused moments pass per snapshot (all columns batched)
    metrics_rows = []
    for j, idf in enumerate(idfs):
        moments = stats_ops.frame_moments(idf, list_of_cols)
        for c in list_of_cols:
            m = moments[c]
            metrics_rows.append(
                {
                    "idx": dfs_count + j,
                    "attribute": c,
                    "type": "Binary" if c in binary_cols else "Numerical",
                    "mean": m.mean,
                    "stddev": m.stddev,
                    "kurtosis": m.kurtosis + 3.0,
                }
            )
    new_df = pd.DataFrame(metrics_rows)

    rows = []
    for c in list_of_cols:
        col_type = "Binary" if c in binary_cols else "Numerical"
        series = new_df[new_df["attribute"] == c][["mean", "stddev", "kurtosis"]]
        if existing_metric_df is not None:
            hist = existing_metric_df[existing_metric_df["attribute"] == c][["mean", "stddev", "kurtosis"]]
            if len(hist):
                series = pd.concat([series, hist], ignore_index=True)
        std_of = series.std(ddof=1)
        mean_of = series.mean()
        rows.append(
            {
                "attribute": c,
                "type": col_type,
                "mean_stddev": std_of["mean"],
                "mean_cv": std_of["mean"] / mean_of["mean"] if mean_of["mean"] else np.nan,
                "stddev_cv": std_of["stddev"] / mean_of["stddev"] if mean_of["stddev"] else np.nan,
                "kurtosis_cv": std_of["kurtosis"] / mean_of["kurtosis"] if mean_of["kurtosis"] else np.nan,
            }
        )
    odf = pd.DataFrame(rows)

    if appended_metric_path:
        app = new_df.copy()
        if existing_metric_df is not None:
            app = pd.concat([existing_metric_df, app], ignore_index=True)
        if dist.rank() == 0:
            if os.path.isdir(appended_metric_path) or "/" in appended_metric_path.rstrip("/"):
                os.makedirs(appended_metric_path, exist_ok=True)
                app.sort_values("idx").to_csv(os.path.join(appended_metric_path, "part-00000.csv"), index=False)
            else:
                app.sort_values("idx").to_csv(appended_metric_path, index=False)
        dist.barrier()

    f_si = compute_si(metric_weightages)
    si_cols = {"mean_si": [], "stddev_si": [], "kurtosis_si": [], "stability_index": []}
    for _, r in odf.iterrows():
        si = f_si(r["type"], r["mean_stddev"], r["mean_cv"], r["stddev_cv"], r["kurtosis_cv"])
        for k, v in zip(si_cols, si):
            si_cols[k].append(v)
    for k, v in si_cols.items():
        odf[k] = v
    odf["flagged"] = [(1 if (s is None or (s == s and s < threshold)) else 0) for s in odf["stability_index"]]
    for c in ["mean_stddev", "mean_cv", "stddev_cv", "kurtosis_cv"]:
        odf[c] = odf[c].round(4)
    if print_impact:
        print("All Attributes:")
        print(odf.to_string(index=False))
        print("Potential Unstable Attributes:")
        print(odf[odf["flagged"] == 1].to_string(index=False))
    return odf


def _read_csv_dir(path):
    import glob

    parts = sorted(glob.glob(os.path.join(path, "*.csv")))
    return pd.concat([pd.read_csv(p) for p in parts], ignore_index=True)


def feature_stability_estimation(
    ctx,
    attribute_stats,
    attribute_transformation,
    metric_weightages={"mean": 0.5, "stddev": 0.3, "kurtosis": 0.2},
    threshold=1,
    print_impact=False,
):
    """Propagate snapshot (mean, stddev) through derived-feature formulas
    via sympy Taylor expansion — reference stability.py:335-589. Returns
    [feature_formula, mean_cv, stddev_cv, mean_si, stddev_si,
    stability_index_lower_bound, stability_index_upper_bound,
    flagged_lower, flagged_upper]."""
    import sympy as sp

    check_metric_weightages(metric_weightages)
    if not isinstance(attribute_stats, pd.DataFrame):
        attribute_stats = attribute_stats.to_pandas()

    def stats_estimation(attributes, transformation, mean, stddev):
        attribute_means = list(zip([sp.Symbol(a) for a in attributes], mean))
        est_mean = 0
        est_var = 0
        expr = sp.parse_expr(transformation)
        for attr, s in zip(attributes, stddev):
            sym = sp.Symbol(attr)
            first_dev = sp.diff(expr, sym)
            second_dev = sp.diff(expr, sym, 2)
            est_mean += s**2 * second_dev.subs(attribute_means) / 2
            est_var += s**2 * (first_dev.subs(attribute_means)) ** 2
        est_mean += expr.subs(attribute_means)
        return [float(est_mean), float(sp.sqrt(est_var))]

    index = sorted(attribute_stats["idx"].unique())
    output = []
    for attrs_key, transformation in attribute_transformation.items():
        attributes = [x.strip() for x in attrs_key.split("|")]
        means_per_idx, stds_per_idx = [], []
        for idx in index:
            attr_mean, attr_std = [], []
            for attr in attributes:
                sub = attribute_stats[(attribute_stats["idx"] == idx) & (attribute_stats["attribute"] == attr)]
                if len(sub) == 0:
                    raise TypeError(
                        "Invalid input for attribute_stats: all involved attributes must have available statistics across all time periods (idx)"
                    )
                attr_mean.append(float(sub["mean"].iloc[0]))
                attr_std.append(float(sub["stddev"].iloc[0]))
            em, es = stats_estimation(attributes, transformation, attr_mean, attr_std)
            means_per_idx.append(em)
            stds_per_idx.append(es)
        mean_cv = round(float(np.std(means_per_idx, ddof=0) / np.mean(means_per_idx)) if np.mean(means_per_idx) else np.nan, 4)
        stddev_cv = round(float(np.std(stds_per_idx, ddof=0) / np.mean(stds_per_idx)) if np.mean(stds_per_idx) else np.nan, 4)
        output.append([transformation, mean_cv, stddev_cv])

    odf = pd.DataFrame(output, columns=["feature_formula", "mean_cv", "stddev_cv"])

    def score_cv(cv, thresholds=[0.03, 0.1, 0.2, 0.5]):
        if cv is None or cv != cv:
            return None
        cv = abs(cv)
        stability_index = [4, 3, 2, 1, 0]
        for i, thresh in enumerate(thresholds):
            if cv < thresh:
                return stability_index[i]
        return stability_index[-1]

    odf["mean_si"] = odf["mean_cv"].map(score_cv)
    odf["stddev_si"] = odf["stddev_cv"].map(score_cv)
    odf["stability_index_lower_bound"] = (
        odf["mean_si"] * metric_weightages.get("mean", 0) + odf["stddev_si"] * metric_weightages.get("stddev", 0)
    ).round(4)
    odf["stability_index_upper_bound"] = (odf["stability_index_lower_bound"] + 4 * metric_weightages.get("kurtosis", 0)).round(4)
    odf["flagged_lower"] = [(1 if (s != s or s < threshold) else 0) for s in odf["stability_index_lower_bound"]]
    odf["flagged_upper"] = [(1 if (s != s or s < threshold) else 0) for s in odf["stability_index_upper_bound"]]
    if print_impact:
        print(odf.to_string(index=False))
    return odf
