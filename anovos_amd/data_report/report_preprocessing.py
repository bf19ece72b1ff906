"""Stats persistence + chart objects for the report (reference parity:
``anovos/data_report/report_preprocessing.py`` :40-735, same on-disk
contract: ``<function_name>.csv`` stats files, plotly JSON charts named
``freqDist_<col>`` / ``eventDist_<col>`` / ``outlier_<col>`` /
``drift_<col>``, plus ``data_type.csv`` under master_path).

MI355X-native: frequency/event-rate tables come from on-device bincount
and two-way label histograms; binning reuses the engine's fused
bucketize kernels via data_transformer.attribute_binning; only tiny
per-bin tables cross to host for plotly.
"""

from __future__ import annotations

import os
from pathlib import Path

import numpy as np
import pandas as pd
import torch

import plotly.express as px
import plotly.graph_objects as go

from anovos_amd.core.dtypes import NULL_CODE
from anovos_amd.core.frame import AnovosFrame, Column
from anovos_amd.data_transformer.transformers import attribute_binning, imputation_MMM, outlier_categories, _load_model
from anovos_amd.shared.utils import attributeType_segregation, ends_with

global_theme = px.colors.sequential.Peach
global_theme_r = px.colors.sequential.Peach_r
global_plot_bg_color = "rgba(0,0,0,0)"
global_paper_bg_color = "rgba(0,0,0,0)"


def save_stats(ctx, idf, master_path, function_name, reread=False, run_type="local",
               mlflow_config=None, auth_key="NA"):
    """Reference report_preprocessing.py:40 — persist a (small) stats
    frame as ``<function_name>.csv`` under master_path; idf is a pandas
    DataFrame or AnovosFrame."""
    from anovos_amd.shared import utils as _su

    # path contract mirrors reference report_preprocessing.py:40-119:
    # local -> master_path directly; databricks -> dbfs:/ resolved to the
    # /dbfs fuse mount; emr/ak8s -> write locally then push via the
    # aws s3 cp / azcopy side channel (shared/utils.cloud_sync)
    if run_type == "local":
        local_path = master_path
    elif run_type == "databricks":
        local_path = _su.output_to_local(master_path)
    else:
        local_path = "report_stats"
    if mlflow_config is not None and mlflow_config.get("track_reports", False):
        local_path = os.path.join(local_path, str(mlflow_config.get("run_id", "run")))
    from anovos_amd.core import dist as _dist

    if _dist.rank() == 0:
        Path(local_path).mkdir(parents=True, exist_ok=True)
        pdf = idf.to_pandas() if isinstance(idf, AnovosFrame) else idf
        pdf.to_csv(ends_with(local_path) + function_name + ".csv", index=False)
        if run_type in ("emr", "ak8s"):
            _su.cloud_sync(ends_with(local_path) + function_name + ".csv",
                           ends_with(master_path) + function_name + ".csv",
                           run_type, auth_key)
    _dist.barrier()
    if reread:
        return pd.read_csv(ends_with(local_path) + function_name + ".csv")
    return None


def edit_binRange(col):
    """Reference report_preprocessing.py:130 — collapse 'a-a' ranges to 'a'."""
    try:
        s = col
        if s is None:
            return s
        parts = str(s).split("-")
        if len(parts) == 2 and parts[0] == parts[1]:
            return parts[0]
        return str(s)
    except Exception:
        return s


def binRange_to_binIdx(ctx, col, cutoffs_path) -> pd.DataFrame:
    """Reference report_preprocessing.py:158 — map bin-range label →
    ordinal bin index using the saved binning model."""
    dfm = _load_model(cutoffs_path, "attribute_binning")
    row = dfm[dfm["attribute"] == col]
    if row.empty:
        return pd.DataFrame(columns=[col, "bin_idx"])
    bc = list(row["parameters"].iloc[0])
    labels = []
    for i in range(len(bc) + 1):
        if i == 0:
            labels.append("<= " + str(round(bc[0], 4)))
        elif i == len(bc):
            labels.append("> " + str(round(bc[-1], 4)))
        else:
            labels.append(str(round(bc[i - 1], 4)) + "-" + str(round(bc[i], 4)))
    labels = [edit_binRange(l) for l in labels]
    return pd.DataFrame({col: labels, "bin_idx": range(len(labels))})


def _freq_table(idf: AnovosFrame, col: str) -> pd.DataFrame:
    """(value, count, count_%) for a categorical/binned column — on-device
    bincount over dictionary codes."""
    c = idf.col(col)
    if c.kind == "categorical":
        codes = c.data.to(torch.long)
        null = codes == NULL_CODE
        n_dict = len(c.dictionary or [])
        cnt = torch.bincount(codes[~null], minlength=n_dict)
        vals = list(c.dictionary or [])
        counts = cnt.cpu().numpy().tolist()
        if bool(null.any()):
            vals.append(None)
            counts.append(int(null.sum()))
    else:
        x = c.data
        null = torch.isnan(x) if x.is_floating_point() else torch.zeros_like(x, dtype=torch.bool)
        uniq, cnts = torch.unique(x[~null], return_counts=True)
        vals = [float(v) for v in uniq.cpu()]
        counts = cnts.cpu().numpy().tolist()
        if bool(null.any()):
            vals.append(None)
            counts.append(int(null.sum()))
    total = max(sum(counts), 1)
    pdf = pd.DataFrame({col: vals, "count": counts})
    pdf = pdf[pdf["count"] > 0]
    pdf["count_%"] = 100.0 * pdf["count"] / total
    return pdf


def plot_frequency(ctx, idf, col, cutoffs_path, is_numcol=False):
    """Reference report_preprocessing.py:200 — frequency bar chart."""
    pdf = _freq_table(idf, col)
    pdf[col] = pdf[col].map(edit_binRange)
    if is_numcol:
        mapping = binRange_to_binIdx(ctx, col, cutoffs_path)
        pdf = pdf.merge(mapping, on=col, how="left").sort_values("bin_idx")
        pdf = pdf.fillna({col: "Missing"})
    else:
        pdf = pdf.sort_values("count", ascending=False).fillna({col: "Missing"})
        pdf.loc[pdf[col] == "others", col] = "others*"
    fig = px.bar(pdf, x=col, y="count",
                 text=pdf["count_%"].apply(lambda x: "{0:1.2f}%".format(x)),
                 color_discrete_sequence=global_theme)
    fig.update_traces(textposition="outside")
    fig.update_layout(title_text=str("Frequency Distribution for " + str(col.upper())))
    fig.update_xaxes(type="category")
    fig.layout.plot_bgcolor = global_plot_bg_color
    fig.layout.paper_bgcolor = global_paper_bg_color
    return fig


def plot_outlier(ctx, idf, col, split_var=None, sample_size=500_000):
    """Reference report_preprocessing.py:260 — violin plot on a bounded
    sample (nulls imputed like the reference via imputation_MMM)."""
    c = idf.col(col)
    x = c.data
    n = x.shape[0]
    if n > sample_size:
        idx = torch.randperm(n, device=x.device)[:sample_size]
        x = x[idx]
    x = x.to(torch.float64)
    med = torch.nanmedian(x) if x.is_floating_point() else x.median()
    x = torch.where(torch.isnan(x), med, x)
    pdf = pd.DataFrame({col: x.cpu().numpy()})
    fig = px.violin(pdf, y=col, box=True, points="outliers",
                    color_discrete_sequence=[global_theme_r[-1], global_theme_r[len(global_theme_r) // 2]])
    fig.layout.plot_bgcolor = global_plot_bg_color
    fig.layout.paper_bgcolor = global_paper_bg_color
    fig.update_layout(legend=dict(orientation="h", x=0.5, yanchor="bottom", xanchor="center"))
    return fig


def _event_rate_table(idf: AnovosFrame, col: str, label_col: str, event_label) -> pd.DataFrame:
    """Per-category 0/1 label counts — one fused two-way bincount."""
    c = idf.col(col)
    lab = idf.col(label_col)
    if lab.kind == "categorical":
        ev_code = None
        for i, s in enumerate(lab.dictionary or []):
            if str(s) == str(event_label):
                ev_code = i
                break
        ev = (lab.data == ev_code) if ev_code is not None else torch.zeros_like(lab.data, dtype=torch.bool)
        lab_null = lab.data == NULL_CODE
    else:
        ev = lab.data == float(event_label)
        lab_null = torch.isnan(lab.data) if lab.data.is_floating_point() else torch.zeros_like(lab.data, dtype=torch.bool)
    codes = c.data.to(torch.long)
    cnull = codes == NULL_CODE if c.kind == "categorical" else (torch.isnan(c.data) if c.data.is_floating_point() else torch.zeros_like(c.data, dtype=torch.bool))
    if c.kind != "categorical":
        uniq, codes = torch.unique(torch.nan_to_num(c.data), return_inverse=True)
        vals = [float(v) for v in uniq.cpu()]
    else:
        vals = list(c.dictionary or [])
    K = len(vals)
    m = ~cnull & ~lab_null
    # single bincount over code*2 + is_event
    packed = codes[m] * 2 + ev[m].to(torch.long)
    cnt = torch.bincount(packed, minlength=2 * K)
    c0 = cnt[0::2].cpu().numpy()
    c1 = cnt[1::2].cpu().numpy()
    pdf = pd.DataFrame({col: vals, "0": c0[:K], "1": c1[:K]})
    pdf = pdf[(pdf["0"] + pdf["1"]) > 0]
    pdf["event_rate"] = 100.0 * pdf["1"] / (pdf["0"] + pdf["1"])
    pdf["attribute_name"] = col
    return pdf


def plot_eventRate(ctx, idf, col, label_col, event_label, cutoffs_path, is_numcol=False):
    """Reference report_preprocessing.py:303 — event-rate by category/bin."""
    pdf = _event_rate_table(idf, col, label_col, event_label)
    pdf[col] = pdf[col].map(edit_binRange)
    if is_numcol:
        mapping = binRange_to_binIdx(ctx, col, cutoffs_path)
        pdf = pdf.merge(mapping, on=col, how="left").sort_values("bin_idx")
    else:
        pdf = pdf.sort_values("event_rate", ascending=False)
        pdf.loc[pdf[col] == "others", col] = "others*"
    fig = px.bar(pdf, x=col, y="event_rate",
                 text=pdf["event_rate"].apply(lambda x: "{0:1.2f}%".format(x)),
                 color_discrete_sequence=global_theme)
    fig.update_traces(textposition="outside")
    fig.update_layout(title_text=str("Event Rate Distribution for " + str(col.upper())
                                     + " [Target Variable : " + str(event_label) + "]"))
    fig.update_xaxes(type="category")
    fig.layout.plot_bgcolor = global_plot_bg_color
    fig.layout.paper_bgcolor = global_paper_bg_color
    return fig


def plot_comparative_drift(ctx, idf, source_pdf, col, cutoffs_path, is_numcol=False):
    """Reference report_preprocessing.py:370 — source vs target frequency
    comparison; source_pdf = saved frequency_counts CSV (cols [col, 'p'])."""
    tgt = _freq_table(idf, col)
    tgt["countpct_target"] = tgt["count"] / max(int(tgt["count"].sum()), 1)
    tgt = tgt[[col, "countpct_target"]]
    src = source_pdf.rename(columns={"p": "countpct_source"})
    src_key = src.columns[0]
    if is_numcol:
        mapping = binRange_to_binIdx(ctx, col, cutoffs_path)
        tgt[col] = tgt[col].map(edit_binRange)
        tgt = tgt.merge(mapping, on=col, how="left")
        src = src.rename(columns={src_key: "bin_idx"})
        pdf = tgt.merge(src[["bin_idx", "countpct_source"]], on="bin_idx", how="outer").sort_values("bin_idx")
    else:
        src = src.rename(columns={src_key: col})
        pdf = tgt.merge(src[[col, "countpct_source"]], on=col, how="outer")
        pdf = pdf.sort_values("countpct_target", ascending=False)
    pdf = pdf.fillna({col: "Missing", "countpct_source": 0, "countpct_target": 0})
    pdf["%_diff"] = (pdf["countpct_target"] / pdf["countpct_source"].replace(0, np.nan) - 1).fillna(0) * 100
    fig = go.Figure()
    fig.add_bar(y=list(pdf.countpct_source.values), x=pdf[col], name="source", marker=dict(color=global_theme))
    fig.update_traces(overwrite=True, marker={"opacity": 0.7})
    fig.add_bar(y=list(pdf.countpct_target.values), x=pdf[col], name="target",
                text=pdf["%_diff"].apply(lambda x: "{0:0.2f}%".format(x)), marker=dict(color=global_theme))
    fig.update_traces(textposition="outside")
    fig.update_layout(paper_bgcolor=global_paper_bg_color, plot_bgcolor=global_plot_bg_color, showlegend=False)
    fig.update_layout(title_text=str("Drift Comparison for " + col + "<br><sup>(L->R : Source->Target)</sup>"))
    fig.update_traces(marker=dict(color=global_theme))
    fig.update_xaxes(type="category")
    fig.update_layout(xaxis_tickfont_size=14, yaxis=dict(title="frequency"))
    return fig



def _write_fig_json(fig, path):
    """All ranks COMPUTE each chart (the underlying stats are
    collectives), but only rank 0 WRITES — concurrent identical writes
    to one path can tear the file."""
    from anovos_amd.core import dist as _dist

    if _dist.rank() == 0:
        fig.write_json(path)


def charts_to_objects(ctx, idf, list_of_cols="all", drop_cols=[], label_col=None,
                      event_label=1, bin_method="equal_range", bin_size=10, coverage=1.0,
                      drift_detector=False, outlier_charts=False, source_path="NA",
                      master_path=".", stats_unique={}, run_type="local", auth_key="NA"):
    """Reference report_preprocessing.py:469 — main driver: bins numeric
    cols (reusing the drift binning model when present), caps categorical
    cardinality, writes one plotly JSON per chart + ``data_type.csv``."""
    num_cols, cat_cols, other_cols = attributeType_segregation(idf)
    if list_of_cols == "all":
        list_of_cols = num_cols + cat_cols
    if isinstance(list_of_cols, str):
        list_of_cols = [x.strip() for x in list_of_cols.split("|")]
    if isinstance(drop_cols, str):
        drop_cols = [x.strip() for x in drop_cols.split("|")]

    from anovos_amd.data_analyzer.stats_generator import uniqueCount_computation

    if stats_unique:
        uc = pd.read_csv(stats_unique["file_path"]) if isinstance(stats_unique, dict) and "file_path" in stats_unique else uniqueCount_computation(ctx, idf, list_of_cols)
    else:
        uc = uniqueCount_computation(ctx, idf, list_of_cols)
    remove_cols = list(uc[uc["unique_values"] < 2]["attribute"])
    list_of_cols = sorted(set(e for e in list_of_cols if e not in (list(drop_cols) + remove_cols)))
    if any(x not in idf.columns for x in list_of_cols) or len(list_of_cols) == 0:
        raise TypeError("Invalid input for Column(s)")
    num_cols, cat_cols, other_cols = attributeType_segregation(idf.select(list_of_cols))

    idf_cleaned = outlier_categories(ctx, idf, list_of_cols=cat_cols, coverage=coverage, max_category=bin_size) if cat_cols else idf
    if source_path == "NA":
        source_path = "intermediate_data"

    if drift_detector:
        try:
            binned_model = _load_model(source_path + "/drift_statistics", "attribute_binning")
            binned_cols = [c for c in list(binned_model["attribute"]) if c in num_cols]
        except Exception:
            binned_cols = []
        to_be_binned = [e for e in num_cols if e not in binned_cols]
    else:
        binned_cols = []
        to_be_binned = num_cols

    idf_encoded = idf_cleaned
    if to_be_binned:
        idf_encoded = attribute_binning(ctx, idf_encoded, list_of_cols=to_be_binned,
                                        method_type=bin_method, bin_size=bin_size,
                                        bin_dtype="categorical", pre_existing_model=False,
                                        model_path=source_path + "/charts_to_objects",
                                        output_mode="append")
    if binned_cols:
        idf_encoded = attribute_binning(ctx, idf_encoded, list_of_cols=binned_cols,
                                        method_type=bin_method, bin_size=bin_size,
                                        bin_dtype="categorical", pre_existing_model=True,
                                        model_path=source_path + "/drift_statistics",
                                        output_mode="append")
    cutoffs_path1 = source_path + "/charts_to_objects"
    cutoffs_path2 = source_path + "/drift_statistics"

    from anovos_amd.shared import utils as _su

    if run_type == "local":
        local_path = master_path
    elif run_type == "databricks":
        local_path = _su.output_to_local(master_path)
    else:
        local_path = "report_stats"
    Path(local_path).mkdir(parents=True, exist_ok=True)

    for col in list_of_cols:
        cutoffs_path = cutoffs_path2 if col in binned_cols else cutoffs_path1
        if col in cat_cols:
            view = idf_encoded
            f = plot_frequency(ctx, view, col, cutoffs_path, is_numcol=False)
            _write_fig_json(f, ends_with(local_path) + "freqDist_" + col)
            if label_col and col != label_col:
                f = plot_eventRate(ctx, view, col, label_col, event_label, cutoffs_path, is_numcol=False)
                _write_fig_json(f, ends_with(local_path) + "eventDist_" + col)
            if drift_detector:
                try:
                    src = pd.read_csv(os.path.join(source_path, "drift_statistics", "frequency_counts", col, "part-00000.csv"))
                    f = plot_comparative_drift(ctx, view, src, col, cutoffs_path, is_numcol=False)
                    _write_fig_json(f, ends_with(local_path) + "drift_" + col)
                except Exception:
                    pass
        if col in num_cols:
            if outlier_charts:
                f = plot_outlier(ctx, idf, col)
                _write_fig_json(f, ends_with(local_path) + "outlier_" + col)
            view = idf_encoded.drop([col]).rename({col + "_binned": col})
            f = plot_frequency(ctx, view, col, cutoffs_path, is_numcol=True)
            _write_fig_json(f, ends_with(local_path) + "freqDist_" + col)
            if label_col and col != label_col:
                f = plot_eventRate(ctx, view, col, label_col, event_label, cutoffs_path, is_numcol=True)
                _write_fig_json(f, ends_with(local_path) + "eventDist_" + col)
            if drift_detector:
                try:
                    src = pd.read_csv(os.path.join(source_path, "drift_statistics", "frequency_counts", col, "part-00000.csv"))
                    f = plot_comparative_drift(ctx, view, src, col, cutoffs_path, is_numcol=True)
                    _write_fig_json(f, ends_with(local_path) + "drift_" + col)
                except Exception:
                    pass

    from anovos_amd.core import dist as _dist

    if _dist.rank() == 0:
        pd.DataFrame(idf.dtypes, columns=["attribute", "data_type"]).to_csv(
            ends_with(local_path) + "data_type.csv", index=False)
    _dist.barrier()
    if run_type in ("emr", "ak8s") and _dist.rank() == 0:
        # push the whole chart-object dir (reference report_preprocessing
        # .py:97-119 recursive aws s3 cp / azcopy side channel)
        _su.cloud_sync(local_path, master_path, run_type, auth_key, recursive=True)
