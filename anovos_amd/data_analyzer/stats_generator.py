"""Descriptive statistics — parity with reference
data_analyzer/stats_generator.py (1,011 LoC).

Every function keeps the reference signature ``f(ctx, idf, list_of_cols,
drop_cols, ..., print_impact)`` and output schema, but the compute path
is MI355X-native: all requested columns are covered by ONE fused moments
kernel launch + batched RCCL all-reduce (ops/stats.py) instead of the
reference's one-Spark-job-per-column loop (stats_generator.py:485-494).
Outputs are small tidy pandas DataFrames (the reference's stats DFs are
all tiny driver-side tables).
"""

from __future__ import annotations

import warnings

import numpy as np
import pandas as pd

from anovos_amd.ops import distinct as distinct_ops
from anovos_amd.ops import groupby as groupby_ops
from anovos_amd.ops import histogram as hist_ops
from anovos_amd.ops import stats as stats_ops
from anovos_amd.shared.utils import attributeType_segregation, normalize_columns
from anovos_amd.shared.tracing import traced


def _r4(x):
    return None if x is None or (isinstance(x, float) and x != x) else round(float(x), 4)


@traced
def global_summary(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[metric, value] — reference stats_generator.py:33-113."""
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if len(cols) == 0:
        raise TypeError("Invalid input for Column(s)")
    row_count = idf.count()
    sub = idf.select(cols)
    num_cols, cat_cols, other_cols = attributeType_segregation(sub)
    rows = [
        ["rows_count", str(row_count)],
        ["columns_count", str(len(cols))],
        ["numcols_count", str(len(num_cols))],
        ["numcols_name", ", ".join(num_cols)],
        ["catcols_count", str(len(cat_cols))],
        ["catcols_name", ", ".join(cat_cols)],
        ["othercols_count", str(len(other_cols))],
        ["othercols_name", ", ".join(other_cols)],
    ]
    odf = pd.DataFrame(rows, columns=["metric", "value"])
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def missingCount_computation(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, missing_count, missing_pct] — reference :116-176."""
    if list_of_cols == "all":
        num_cols, cat_cols, _ = attributeType_segregation(idf)
        list_of_cols = num_cols + cat_cols
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if len(cols) == 0:
        raise TypeError("Invalid input for Column(s)")
    nulls, total = stats_ops.null_counts(idf, cols)
    odf = pd.DataFrame(
        {
            "attribute": cols,
            "missing_count": [nulls[c] for c in cols],
            "missing_pct": [_r4(nulls[c] / total) if total else None for c in cols],
        }
    )
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def nonzeroCount_computation(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, nonzero_count, nonzero_pct] — reference :179-248.
    Computed from the fused moments pass (zero_count slot)."""
    num_all = attributeType_segregation(idf)[0]
    if list_of_cols == "all":
        list_of_cols = num_all
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if len(cols) == 0:
        warnings.warn("No Non-Zero Count Computation - No numerical column(s) to analyze")
        return pd.DataFrame(columns=["attribute", "nonzero_count", "nonzero_pct"])
    moments = stats_ops.frame_moments(idf, cols)
    total = idf.count()
    nz = [int(moments[c].n - moments[c].zeros) for c in cols]
    odf = pd.DataFrame(
        {
            "attribute": cols,
            "nonzero_count": nz,
            "nonzero_pct": [_r4(v / total) if total else None for v in nz],
        }
    )
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def measures_of_counts(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, fill_count, fill_pct, missing_count, missing_pct,
    nonzero_count, nonzero_pct] — reference :251-325."""
    if list_of_cols == "all":
        num_cols, cat_cols, _ = attributeType_segregation(idf)
        list_of_cols = num_cols + cat_cols
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if len(cols) == 0:
        raise TypeError("Invalid input for Column(s)")
    num_cols = attributeType_segregation(idf.select(cols))[0]
    nulls, total = stats_ops.null_counts(idf, cols)
    miss = pd.DataFrame(
        {
            "attribute": cols,
            "fill_count": [total - nulls[c] for c in cols],
            "fill_pct": [_r4((total - nulls[c]) / total) if total else None for c in cols],
            "missing_count": [nulls[c] for c in cols],
            "missing_pct": [_r4(nulls[c] / total) if total else None for c in cols],
        }
    )
    nz = nonzeroCount_computation(ctx, idf, num_cols) if num_cols else pd.DataFrame(columns=["attribute", "nonzero_count", "nonzero_pct"])
    odf = miss.merge(nz, on="attribute", how="outer")
    odf = odf.set_index("attribute").loc[[c for c in cols]].reset_index()
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def mode_computation(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, mode, mode_rows] — reference :328-421 (groupBy-count
    top-1 per column; here a fused dictionary bincount / exact unique)."""
    if list_of_cols == "all":
        num_cols, cat_cols, _ = attributeType_segregation(idf)
        list_of_cols = num_cols + cat_cols
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if len(cols) == 0:
        warnings.warn("No Mode Computation - No discrete column(s) to analyze")
        return pd.DataFrame(columns=["attribute", "mode", "mode_rows"])
    cat_cols = [c for c in cols if idf.col(c).kind == "categorical"]
    num_cols = [c for c in cols if idf.col(c).kind == "numerical"]
    counts = groupby_ops.cat_value_counts(idf, cat_cols) if cat_cols else {}
    num_modes = groupby_ops.discrete_modes(idf, num_cols) if num_cols else {}
    rows = []
    for c in cols:
        if c in num_modes:
            m, n = num_modes[c]
        else:
            m, n = groupby_ops.mode(idf, c, counts.get(c))
        rows.append([c, None if m is None else str(m), n])
    odf = pd.DataFrame(rows, columns=["attribute", "mode", "mode_rows"])
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def measures_of_centralTendency(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, mean, median, mode, mode_rows, mode_pct] — reference
    :424-526. mean/median only for numeric columns."""
    num_all, cat_all, _ = attributeType_segregation(idf)
    if list_of_cols == "all":
        list_of_cols = num_all + cat_all
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if len(cols) == 0:
        raise TypeError("Invalid input for Column(s)")
    num_cols = [c for c in cols if c in num_all]
    moments = stats_ops.frame_moments(idf, num_cols) if num_cols else {}
    medians = hist_ops.approx_quantiles(idf, num_cols, [0.5], moments=moments) if num_cols else {}
    nulls, total = stats_ops.null_counts(idf, cols)
    dfm = mode_computation(ctx, idf, cols)
    mode_map = dict(zip(dfm["attribute"], zip(dfm["mode"], dfm["mode_rows"])))
    rows = []
    for c in cols:
        fill = total - nulls[c]
        mode_v, mode_rows = mode_map.get(c, (None, None))
        rows.append(
            [
                c,
                _r4(moments[c].mean) if c in moments else None,
                _r4(medians[c][0]) if c in medians else None,
                mode_v,
                mode_rows,
                _r4(mode_rows / fill) if mode_rows is not None and fill else None,
            ]
        )
    odf = pd.DataFrame(rows, columns=["attribute", "mean", "median", "mode", "mode_rows", "mode_pct"])
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def uniqueCount_computation(
    ctx, idf, list_of_cols="all", drop_cols=[], compute_approx_unique_count=False, rsd=None, print_impact=False
):
    """[attribute, unique_values] — reference :529-620. Approx path is
    the HLL kernel (all-reduce(max) register merge)."""
    if list_of_cols == "all":
        num_cols, cat_cols, _ = attributeType_segregation(idf)
        list_of_cols = num_cols + cat_cols
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if rsd is not None and rsd < 0:
        raise ValueError("rsd value can not be less than 0 (default value is 0.05)")
    if len(cols) == 0:
        warnings.warn("No Unique Count Computation - No discrete column(s) to analyze")
        return pd.DataFrame(columns=["attribute", "unique_values"])
    if compute_approx_unique_count:
        num_cols = [c for c in cols if idf.col(c).kind == "numerical"]
        other = [c for c in cols if c not in num_cols]
        vals = distinct_ops.approx_distinct(idf, num_cols) if num_cols else {}
        vals.update(distinct_ops.exact_distinct(idf, other) if other else {})
    else:
        vals = distinct_ops.exact_distinct(idf, cols)
    odf = pd.DataFrame({"attribute": cols, "unique_values": [vals[c] for c in cols]})
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def measures_of_cardinality(
    ctx, idf, list_of_cols="all", drop_cols=[], use_approx_unique_count=True, rsd=0.05, print_impact=False
):
    """[attribute, unique_values, IDness] — reference :623-733.
    IDness = unique/(rows - missing)."""
    if list_of_cols == "all":
        num_cols, cat_cols, _ = attributeType_segregation(idf)
        list_of_cols = num_cols + cat_cols
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if rsd is not None and rsd < 0:
        raise ValueError("rsd value can not be less than 0 (default value is 0.05)")
    if len(cols) == 0:
        warnings.warn("No Cardinality Computation - No discrete column(s) to analyze")
        return pd.DataFrame(columns=["attribute", "unique_values", "IDness"])
    uc = uniqueCount_computation(ctx, idf, cols, compute_approx_unique_count=use_approx_unique_count, rsd=rsd)
    nulls, total = stats_ops.null_counts(idf, cols)
    uc = uc.set_index("attribute")
    rows = []
    for c in cols:
        u = uc.loc[c, "unique_values"]
        denom = total - nulls[c]
        rows.append([c, u, _r4(min(u / denom, 1.0)) if denom else None])
    odf = pd.DataFrame(rows, columns=["attribute", "unique_values", "IDness"])
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def measures_of_dispersion(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, stddev, variance, cov, IQR, range] — reference :736-829."""
    num_all = attributeType_segregation(idf)[0]
    if list_of_cols == "all":
        list_of_cols = num_all
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if any(x not in num_all for x in cols):
        raise TypeError("Invalid input for Column(s)")
    if len(cols) == 0:
        warnings.warn("No Dispersion Computation - No numerical column(s) to analyze")
        return pd.DataFrame(columns=["attribute", "stddev", "variance", "cov", "IQR", "range"])
    moments = stats_ops.frame_moments(idf, cols)
    quarts = hist_ops.approx_quantiles(idf, cols, [0.25, 0.75], moments=moments)
    rows = []
    for c in cols:
        m = moments[c]
        sd = m.stddev
        sd_r = _r4(sd)
        rows.append(
            [
                c,
                sd_r,
                _r4(sd_r * sd_r) if sd_r is not None else None,
                _r4(sd_r / m.mean) if sd_r is not None and m.mean not in (0,) and m.mean == m.mean else None,
                _r4(quarts[c][1] - quarts[c][0]) if quarts[c][0] == quarts[c][0] else None,
                _r4(m.max - m.min) if m.max == m.max else None,
            ]
        )
    odf = pd.DataFrame(rows, columns=["attribute", "stddev", "variance", "cov", "IQR", "range"])
    if print_impact:
        print(odf.to_string(index=False))
    return odf


PERCENTILE_STATS = ["min", "1%", "5%", "10%", "25%", "50%", "75%", "90%", "95%", "99%", "max"]
_PROBS = [0.01, 0.05, 0.10, 0.25, 0.50, 0.75, 0.90, 0.95, 0.99]


@traced
def measures_of_percentiles(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, min, 1%..99%, max] — reference :832-916."""
    num_all = attributeType_segregation(idf)[0]
    if list_of_cols == "all":
        list_of_cols = num_all
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if any(x not in num_all for x in cols):
        raise TypeError("Invalid input for Column(s)")
    if len(cols) == 0:
        warnings.warn("No Percentiles Computation - No numerical column(s) to analyze")
        return pd.DataFrame(columns=["attribute"] + PERCENTILE_STATS)
    moments = stats_ops.frame_moments(idf, cols)
    q = hist_ops.approx_quantiles(idf, cols, _PROBS, moments=moments)
    rows = []
    for c in cols:
        m = moments[c]
        rows.append([c, _r4(m.min)] + [_r4(v) for v in q[c]] + [_r4(m.max)])
    odf = pd.DataFrame(rows, columns=["attribute"] + PERCENTILE_STATS)
    if print_impact:
        print(odf.to_string(index=False))
    return odf


@traced
def measures_of_shape(ctx, idf, list_of_cols="all", drop_cols=[], print_impact=False):
    """[attribute, skewness, kurtosis] — reference :919-1011 (F.skewness /
    F.kurtosis: population skew, excess kurtosis)."""
    num_all = attributeType_segregation(idf)[0]
    if list_of_cols == "all":
        list_of_cols = num_all
    cols = normalize_columns(idf, list_of_cols, drop_cols)
    if any(x not in num_all for x in cols):
        raise TypeError("Invalid input for Column(s)")
    if len(cols) == 0:
        warnings.warn("No Skewness/Kurtosis Computation - No numerical column(s) to analyze")
        return pd.DataFrame(columns=["attribute", "skewness", "kurtosis"])
    moments = stats_ops.frame_moments(idf, cols)
    rows = [[c, _r4(moments[c].skewness), _r4(moments[c].kurtosis)] for c in cols]
    odf = pd.DataFrame(rows, columns=["attribute", "skewness", "kurtosis"])
    if print_impact:
        print(odf.to_string(index=False))
    return odf
