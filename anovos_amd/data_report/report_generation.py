"""Final HTML report assembly (reference parity:
``anovos/data_report/report_generation.py`` :3984-4413 — reads only
*files on disk* under master_path: stats CSVs written by save_stats and
plotly JSON chart objects written by charts_to_objects, then emits
``ml_anovos_report.html``).

The reference assembled tabs with datapane; that package is not in this
stack, so the same tab structure (executive summary, wiki, descriptive
statistics, quality check, attribute associations, drift & stability,
time series, geospatial) is rendered by a small native HTML/CSS/JS
template with plotly.js inlined — the report is a single offline file,
as before. Chart JSONs and CSV names are consumed under the exact
reference contract (report_generation.py:4111-4245), so artifacts
produced by either engine interchange.
"""

from __future__ import annotations

import html as _html
import json
import os
from typing import Dict, List, Optional

import numpy as np
import pandas as pd

import plotly.graph_objects as go
import plotly.io as pio

from anovos_amd.shared.utils import ends_with


def remove_u_score(col: str) -> str:
    """`_`-separated name -> title-cased words for display labels
    (reference report_generation.py:78)."""
    return " ".join(w.title() for w in str(col).split("_"))


SG_tabs = [
    "measures_of_counts", "measures_of_centralTendency", "measures_of_cardinality",
    "measures_of_percentiles", "measures_of_dispersion", "measures_of_shape", "global_summary",
]
QC_tabs = [
    "nullColumns_detection", "IDness_detection", "biasedness_detection",
    "invalidEntries_detection", "duplicate_detection", "nullRows_detection", "outlier_detection",
]
AE_tabs = ["correlation_matrix", "IV_calculation", "IG_calculation", "variable_clustering"]
drift_tab = ["drift_statistics"]
stability_tab = ["stability_index", "stabilityIndex_metrics"]

STABILITY_INTERPRETATION = pd.DataFrame(
    [["0-1", "Very Unstable"], ["1-2", "Unstable"], ["2-3", "Marginally Stable"],
     ["3-3.5", "Stable"], ["3.5-4", "Very Stable"]],
    columns=["StabilityIndex", "StabilityOrder"],
)

_CSS = """
body{font-family:'Segoe UI',Roboto,Helvetica,Arial,sans-serif;margin:0;background:#fafafa;color:#222}
header{background:#273746;color:#fff;padding:18px 32px}
header h1{margin:0;font-size:22px} header p{margin:4px 0 0;font-size:13px;color:#aeb6bf}
.tabbar{display:flex;flex-wrap:wrap;background:#1f2e3d;padding:0 24px}
.tabbar button{background:none;border:none;color:#d5dbdb;padding:12px 18px;font-size:14px;cursor:pointer;border-bottom:3px solid transparent}
.tabbar button.active{color:#fff;border-bottom-color:#f5b041;font-weight:600}
.tab{display:none;padding:24px 32px}.tab.active{display:block}
h2{font-size:18px;border-bottom:2px solid #e5e8e8;padding-bottom:6px}
h3{font-size:15px;color:#34495e;margin-top:28px}
table.stats{border-collapse:collapse;font-size:12.5px;margin:8px 0;background:#fff}
table.stats th{background:#273746;color:#fff;padding:6px 10px;text-align:left}
table.stats td{border:1px solid #e5e8e8;padding:5px 10px}
table.stats tr:nth-child(even){background:#f4f6f6}
.grid{display:flex;flex-wrap:wrap;gap:16px}.grid>div{flex:1 1 540px;background:#fff;border:1px solid #eee;border-radius:6px;padding:6px}
.kpi{display:inline-block;background:#fff;border:1px solid #e5e8e8;border-radius:8px;padding:14px 26px;margin:6px;text-align:center}
.kpi .v{font-size:26px;font-weight:700;color:#273746}.kpi .l{font-size:12px;color:#7b8a8b}
.note{font-size:12px;color:#7b8a8b}
"""

_JS = """
function showTab(i){
  var tabs=document.querySelectorAll('.tab');var btns=document.querySelectorAll('.tabbar button');
  tabs.forEach(function(t,j){t.classList.toggle('active',i===j)});
  btns.forEach(function(b,j){b.classList.toggle('active',i===j)});
  window.dispatchEvent(new Event('resize'));
}
"""


def _tbl(pdf: pd.DataFrame, max_rows: int = 200) -> str:
    if pdf is None or len(pdf) == 0:
        return "<p class='note'>No data.</p>"
    p = pdf.head(max_rows).copy()
    return p.to_html(index=False, classes="stats", border=0, float_format=lambda v: f"{v:.4f}")


def _fig_div(fig_json_path_or_fig, div_id: str) -> str:
    if isinstance(fig_json_path_or_fig, str):
        with open(fig_json_path_or_fig) as f:
            spec = json.load(f)
    else:
        spec = json.loads(pio.to_json(fig_json_path_or_fig))
    data = json.dumps(spec.get("data", []))
    layout = json.dumps(spec.get("layout", {}))
    return (f"<div id='{div_id}' class='plt'></div>"
            f"<script>Plotly.newPlot('{div_id}',{data},{layout},{{responsive:true}});</script>")


def _read_csv(master_path: str, name: str) -> Optional[pd.DataFrame]:
    p = ends_with(master_path) + name + ".csv"
    if os.path.exists(p):
        try:
            return pd.read_csv(p)
        except Exception:
            return None
    return None


def _chart_files(master_path: str, prefix: str) -> List[str]:
    return sorted(x for x in os.listdir(master_path)
                  if x.startswith(prefix) and not x.endswith(".csv") and not x.endswith(".html"))


def _charts_section(master_path: str, files: List[str], title: str, uid: str, limit: int = 60) -> str:
    if not files:
        return ""
    out = [f"<h3>{_html.escape(title)}</h3><div class='grid'>"]
    for i, fn in enumerate(files[:limit]):
        try:
            out.append("<div>" + _fig_div(os.path.join(master_path, fn), f"{uid}_{i}") + "</div>")
        except Exception:
            continue
    out.append("</div>")
    return "".join(out)


def _diagnosis_matrix(master_path: str, corr_threshold, iv_threshold) -> Optional[pd.DataFrame]:
    """The reference's data-diagnosis matrix (report_generation.py:617-790):
    attribute x {High Variance, Positive/Negative Skewness, High/Low
    Kurtosis, Low Fill Rates, High Biasedness, Outliers, High
    Correlation, Significant Attributes} with check/cross marks."""
    import numpy as np

    checks = []

    def q(name, expr, metric):
        df = _read_csv(master_path, name)
        try:
            vals = list(df.query(expr)["attribute"].values) if df is not None else []
        except Exception:
            vals = []
        checks.append((metric, vals))

    q("measures_of_dispersion", "`cov`>1", "High Variance")
    q("measures_of_shape", "`skewness`>0", "Positive Skewness")
    q("measures_of_shape", "`skewness`<0", "Negative Skewness")
    q("measures_of_shape", "`kurtosis`>0", "High Kurtosis")
    q("measures_of_shape", "`kurtosis`<0", "Low Kurtosis")
    q("measures_of_counts", "`fill_pct`<0.7", "Low Fill Rates")
    bd = _read_csv(master_path, "biasedness_detection")
    try:
        col = "treated" if (bd is not None and "treated" in bd.columns) else "flagged"
        checks.append(("High Biasedness", list(bd.query(f"`{col}`>0")["attribute"].values) if bd is not None else []))
    except Exception:
        checks.append(("High Biasedness", []))
    od = _read_csv(master_path, "outlier_detection")
    checks.append(("Outliers", list(od["attribute"].values) if od is not None else []))
    corr = _read_csv(master_path, "correlation_matrix")
    hi_corr = []
    if corr is not None and "attribute" in corr.columns:
        try:
            mat = corr[list(corr["attribute"].values)]
            upper = mat.where(np.triu(np.ones(mat.shape), k=1).astype(bool))
            hi_corr = [c for c in upper.columns if (upper[c] > corr_threshold).any()]
        except Exception:
            hi_corr = []
    checks.append(("High Correlation", hi_corr))
    iv = _read_csv(master_path, "IV_calculation")
    try:
        checks.append(("Significant Attributes",
                       list(iv.query(f"`iv`>{iv_threshold}")["attribute"].values) if iv is not None else []))
    except Exception:
        checks.append(("Significant Attributes", []))

    attrs = sorted({a for _, vals in checks for a in vals})
    if not attrs:
        return None
    # every attribute seen anywhere gets a row; metric column order is
    # the reference's (report_generation.py:773-787)
    order = ["Outliers", "Significant Attributes", "Positive Skewness", "Negative Skewness",
             "High Variance", "High Correlation", "High Kurtosis", "Low Kurtosis"]
    rows = []
    bymetric = dict(checks)
    for a in attrs:
        rows.append([a] + [("✔" if a in bymetric.get(mname, []) else "✘") for mname in order])
    return pd.DataFrame(rows, columns=["Attribute"] + order)


def executive_summary_gen(master_path: str, label_col, event_label,
                          corr_threshold=0.4, iv_threshold=0.02, drift_threshold_model=0.1,
                          ds_ind=None, id_col=None, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:524-908 — narrative summary, label
    distribution pie, the data-diagnosis matrix, and the drift/stability
    health BigNumbers."""
    gs = _read_csv(master_path, "global_summary")
    if gs is None:
        return "<p class='note'>global_summary.csv not found.</p>"
    get = lambda m: gs[gs["metric"] == m]["value"].values
    kpis = []
    for m, label in (("rows_count", "Rows"), ("columns_count", "Columns"),
                     ("numcols_count", "Numerical Columns"), ("catcols_count", "Categorical Columns")):
        v = get(m)
        kpis.append(f"<div class='kpi'><div class='v'>{_html.escape(str(v[0])) if len(v) else '—'}</div><div class='l'>{label}</div></div>")
    # narrative line (reference :585-595)
    def _num(m):
        v = get(m)
        try:
            return int(float(v[0]))
        except Exception:
            return 0
    narrative = (f"<p><b>Key Report Highlights</b></p><p>The dataset contains <b>{_num('rows_count'):,}</b> "
                 f"records and <b>{_num('numcols_count') + _num('catcols_count')}</b> attributes "
                 f"(<b>{_num('numcols_count')}</b> numerical + <b>{_num('catcols_count')}</b> categorical).</p>")
    lab = (f"<p>Target variable is <b>{_html.escape(str(label_col))}</b> "
           f"(event label: <b>{_html.escape(str(event_label))}</b>)</p>"
           if label_col else "<p>There is <b>no</b> target variable in the dataset</p>")
    extra = ""
    if label_col:
        # label distribution pie from the label's frequency chart object
        # (reference :556-580 rebuilds a pie from freqDist_<label>)
        p_lab = ends_with(master_path) + "freqDist_" + str(label_col)
        if os.path.exists(p_lab):
            try:
                obj = json.load(open(p_lab))
                tr = obj["data"][0] if isinstance(obj, dict) and "data" in obj else None
                if tr is not None and "x" in tr and "y" in tr:
                    pie = go.Figure(go.Pie(labels=tr["x"], values=tr["y"], textinfo="label+percent",
                                           insidetextorientation="radial", pull=[0, 0.1]))
                    pie.update_traces(textposition="inside", textinfo="percent+label")
                    pie.update_layout(height=360, legend=dict(orientation="h", x=0.5, xanchor="center"))
                    extra = "<h3>Label Distribution</h3>" + _fig_div(pie, "exec_label")
                else:
                    extra = "<h3>Label Distribution</h3>" + _fig_div(p_lab, "exec_label")
            except Exception:
                extra = ""
    diag = _diagnosis_matrix(master_path, corr_threshold, iv_threshold)
    diag_html = ""
    if diag is not None:
        diag_html = "<h3>Data Diagnosis</h3>" + _tbl(diag, 500)
    # drift / stability health BigNumbers (reference :793-860)
    flags = []
    dd = _read_csv(master_path, "drift_statistics")
    if dd is not None and "flagged" in dd.columns:
        v = pd.to_numeric(dd["flagged"], errors="coerce").fillna(0)
        n, tot = int(v.sum()), int(len(v))
        pct = round(100.0 * n / max(tot, 1), 2)
        flags.append(f"<div class='kpi'><div class='v'>{n} / {tot}</div><div class='l'># Drifted Attributes</div></div>")
        flags.append(f"<div class='kpi'><div class='v'>{pct}%</div><div class='l'>% Drifted Attributes</div></div>")
    si = _read_csv(master_path, "stability_index")
    if si is not None and "flagged" in si.columns:
        v = pd.to_numeric(si["flagged"], errors="coerce").fillna(0)
        n, tot = int((v > 0).sum()), int(len(v))
        pct = round(100.0 * n / max(tot, 1), 2)
        flags.append(f"<div class='kpi'><div class='v'>{n} / {tot}</div><div class='l'># Unstable Attributes</div></div>")
        flags.append(f"<div class='kpi'><div class='v'>{pct}%</div><div class='l'>% Unstable Attributes</div></div>")
    health = ("<h3>Data Health (Drift &amp; Stability)</h3>" + "".join(flags)) if flags else ""
    return "".join(kpis) + narrative + lab + extra + diag_html + health


_METRIC_DICT = [
    ("fill_count / fill_pct", "rows with a non-null value for the attribute, and their share of all rows"),
    ("missing_count / missing_pct", "rows with a null value, and their share of all rows"),
    ("nonzero_count / nonzero_pct", "rows with a value different from zero (numerical attributes)"),
    ("mean / median", "arithmetic mean and 50th percentile of the non-null values"),
    ("mode / mode_rows / mode_pct", "most frequent value, its row count and its share of non-null rows"),
    ("unique_values", "number of distinct non-null values (HyperLogLog approximation at scale)"),
    ("IDness", "unique_values divided by non-null rows — 1.0 marks identifier-like attributes"),
    ("stddev / variance / cov", "sample standard deviation, variance and coefficient of variation"),
    ("IQR", "inter-quartile range (75th minus 25th percentile)"),
    ("skewness / kurtosis", "third / fourth standardized moments (population, excess kurtosis)"),
    ("PSI", "population stability index between source and target bin frequencies"),
    ("JSD", "Jensen-Shannon divergence between source and target distributions"),
    ("HD", "Hellinger distance between source and target distributions"),
    ("KS", "Kolmogorov-Smirnov statistic: max CDF gap between source and target"),
    ("stability_index", "weighted CV-derived 0-4 score of an attribute across snapshots (4 = most stable)"),
    ("IV", "information value of an attribute against the binary label (WOE-weighted)"),
    ("IG", "information gain: label entropy reduction from splitting on the attribute"),
]


def wiki_generator(master_path: str, dataDict_path=None, metricDict_path=None, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:909 — data dictionary + metric
    definitions tab. Falls back to the engine's built-in metric
    definitions when no metricDict CSV is supplied."""
    parts = ["<p><i>A quick reference to the attributes from the dataset (Data "
             "Dictionary) and the metrics computed in the report (Metric "
             "Dictionary).</i></p>"]
    # data dictionary OUTER-merged with the observed data types
    # (reference :931-948: dataDict ⟗ data_type.csv on attribute)
    datatype_df = _read_csv(master_path, "data_type")
    dd_df = None
    if dataDict_path and os.path.exists(str(dataDict_path)):
        try:
            dd_df = pd.read_csv(dataDict_path)
            dd_df.columns = [c.strip().lower() if c.strip().lower() == "attribute" else c for c in dd_df.columns]
            if "attribute" not in dd_df.columns and len(dd_df.columns):
                dd_df = dd_df.rename(columns={dd_df.columns[0]: "attribute"})
        except Exception:
            dd_df = None
    if dd_df is not None and datatype_df is not None:
        try:
            merged = dd_df.merge(datatype_df, how="outer", on="attribute")
        except Exception:
            merged = dd_df
        parts.append("<h3>Data Dictionary</h3>" + _tbl(merged, 500))
    elif dd_df is not None:
        parts.append("<h3>Data Dictionary</h3>" + _tbl(dd_df, 500))
    elif datatype_df is not None:
        parts.append("<h3>Data Dictionary</h3>" + _tbl(datatype_df, 500))
    if metricDict_path and os.path.exists(metricDict_path):
        try:
            parts.append("<h3>Metric Dictionary</h3>" + _tbl(pd.read_csv(metricDict_path), 500))
        except Exception:
            pass
    else:
        # packaged default: the reference's full 89-row metric table
        builtin = os.path.join(os.path.dirname(os.path.abspath(__file__)), "data", "metric_dictionary.csv")
        try:
            parts.append("<h3>Metric Dictionary</h3>" + _tbl(pd.read_csv(builtin), 500))
        except Exception:
            parts.append("<h3>Metric Dictionary</h3>"
                         + _tbl(pd.DataFrame(_METRIC_DICT, columns=["metric", "definition"]), 100))
    return "".join(parts)


def _split_charts_by_kind(master_path: str, prefix: str):
    """Split chart objects into numerical / categorical grids using
    data_type.csv (reference descriptive_statistics keeps separate
    all_charts_num_ / all_charts_cat_ grids)."""
    files = _chart_files(master_path, prefix)
    dt = _read_csv(master_path, "data_type")
    if dt is None:
        return files, []
    kind = {str(a): str(t) for a, t in zip(dt["attribute"], dt["data_type"])}
    num, cat = [], []
    for fn in files:
        attr = fn[len(prefix):]
        t = kind.get(attr, "")
        (cat if t in ("string", "categorical") else num).append(fn)
    return num, cat


def descriptive_statistics(master_path: str, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:994-1153 — global summary
    narrative, per-metric stat tables, numerical/categorical chart
    grids."""
    parts = []
    gs = _read_csv(master_path, "global_summary")
    if gs is not None:
        # the reference renders global_summary as a bullet list
        items = "".join(f"<li><b>{_html.escape(str(m))}</b>: {_html.escape(str(v))}</li>"
                        for m, v in zip(gs["metric"], gs["value"]))
        parts.append(f"<h3>Global Summary</h3><ul>{items}</ul>")
    for name in SG_tabs:
        df = _read_csv(master_path, name)
        if df is not None:
            parts.append(f"<h3>{remove_u_score(name)}</h3>" + _tbl(df))
    num, cat = _split_charts_by_kind(master_path, "freqDist_")
    parts.append(_charts_section(master_path, num, "Numerical Attribute Distributions", "fdn"))
    parts.append(_charts_section(master_path, cat, "Categorical Attribute Frequencies", "fdc"))
    return "".join(parts) if parts else "<p class='note'>No descriptive statistics saved.</p>"


_QC_ROW_LEVEL = ["duplicate_detection", "nullRows_detection"]


def quality_check(master_path: str, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:1154-1290 — row-level and
    column-level sub-sections + outlier violin charts."""
    row_parts, col_parts = [], []
    for name in QC_tabs:
        df = _read_csv(master_path, name)
        if df is not None:
            (row_parts if name in _QC_ROW_LEVEL else col_parts).append(
                f"<h3>{remove_u_score(name)}</h3>" + _tbl(df))
    parts = []
    if row_parts:
        parts.append("<h2>Row-Level Checks</h2>" + "".join(row_parts))
    if col_parts:
        parts.append("<h2>Column-Level Checks</h2>" + "".join(col_parts))
    parts.append(_charts_section(master_path, _chart_files(master_path, "outlier_"), "Outlier Charts", "oc"))
    return "".join(parts) if parts else "<p class='note'>No quality-check statistics saved.</p>"


def attribute_associations(master_path: str, label_col, event_label, corr_threshold=0.4, iv_threshold=0.02, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:1291 — correlation heatmap, IV/IG
    bars, variable clustering table + event-rate charts."""
    parts = []
    corr = _read_csv(master_path, "correlation_matrix")
    if corr is not None:
        attrs = list(corr.get("attribute", corr.columns))
        mat = corr.drop(columns=["attribute"], errors="ignore")
        fig = go.Figure(go.Heatmap(z=mat.values, x=list(mat.columns), y=attrs, colorscale="Peach", zmin=-1, zmax=1))
        fig.update_layout(title="Correlation Matrix", height=520)
        if corr_threshold:
            import numpy as _np

            v = _np.abs(mat.to_numpy(dtype=float))
            _np.fill_diagonal(v, 0.0)
            n_pairs = int((v >= corr_threshold).sum() // 2)
            parts.append(f"<div class='kpi'><div class='v'>{n_pairs}</div>"
                         f"<div class='l'>attribute pairs with |corr| ≥ {corr_threshold}</div></div>")
        parts.append("<h3>Correlation Matrix</h3>" + _fig_div(fig, "corrheat"))
    for name, metric in (("IV_calculation", "iv"), ("IG_calculation", "ig")):
        df = _read_csv(master_path, name)
        if df is not None:
            vcol = [c for c in df.columns if c != "attribute"][0]
            fig = go.Figure(go.Bar(x=df["attribute"], y=df[vcol]))
            if metric == "iv" and iv_threshold:
                fig.add_hline(y=iv_threshold, line_dash="dash",
                              annotation_text=f"iv_threshold={iv_threshold}")
                n_pred = int((pd.to_numeric(df[vcol], errors="coerce") >= iv_threshold).sum())
                parts.append(f"<div class='kpi'><div class='v'>{n_pred}</div>"
                             f"<div class='l'>attributes with IV ≥ {iv_threshold}</div></div>")
            fig.update_layout(title=name, height=380)
            parts.append(f"<h3>{name}</h3>" + _fig_div(fig, f"bar_{metric}") + _tbl(df))
    vc = _read_csv(master_path, "variable_clustering")
    if vc is not None:
        parts.append("<h3>Variable Clustering</h3>" + _tbl(vc))
    parts.append(_charts_section(master_path, _chart_files(master_path, "eventDist_"), "Event-Rate Distributions", "ev"))
    return "".join(parts) if parts else "<p class='note'>No association statistics saved.</p>"


def data_drift_stability(master_path: str, drift_threshold_model=0.1, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:1434."""
    parts = []
    dd = _read_csv(master_path, "drift_statistics")
    if dd is not None:
        parts.append("<h3>Drift Statistics</h3>" + _tbl(dd))
        if "flagged" in dd.columns:
            n_drift = int(pd.to_numeric(dd["flagged"], errors="coerce").fillna(0).sum())
            parts.insert(0, f"<div class='kpi'><div class='v'>{n_drift}</div><div class='l'>Drifted Attributes</div></div>")
    parts.append(_charts_section(master_path, [x for x in _chart_files(master_path, "drift_") if x != "drift_statistics"], "Source vs Target Distributions", "dr"))
    si = _read_csv(master_path, "stability_index")
    si_metrics = _read_csv(master_path, "stabilityIndex_metrics")
    if si is None:
        si = si_metrics
    if si is not None:
        parts.append("<h3>Stability Index</h3>" + _tbl(si))
        # per-attribute metric trajectories across snapshots (reference
        # line_chart_gen_stability grid, report_generation.py:99)
        if si_metrics is not None and "attribute" in si_metrics.columns:
            charts = []
            for col in list(dict.fromkeys(si_metrics["attribute"]))[:40]:
                try:
                    charts.append("<div>" + line_chart_gen_stability(si, si_metrics, col) + "</div>")
                except Exception:
                    continue
            if charts:
                parts.append("<h3>Stability Trajectories</h3><div class='grid'>" + "".join(charts) + "</div>")
        parts.append("<h3>Stability Interpretation</h3>" + _tbl(STABILITY_INTERPRETATION))
    return "".join(parts) if parts else "<p class='note'>No drift / stability statistics saved.</p>"


def ts_viz_generate(master_path: str, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:3091 — time-series tab from
    ts_analyzer CSVs (stats_<col>_{1,2}.csv + <ts>_<attr>_<type>.csv)."""
    files = [x for x in os.listdir(master_path) if x.startswith("stats_") and x.endswith(".csv")]
    if not files and not os.path.exists(ends_with(master_path) + "ts_cols_stats.csv"):
        return ""
    parts = []
    tcs = _read_csv(master_path, "ts_cols_stats")
    if tcs is not None:
        parts.append("<h3>Detected Timestamp Columns</h3>" + _tbl(tcs))
    for fn in sorted(files):
        try:
            parts.append(f"<h3>{fn[:-4]}</h3>" + _tbl(pd.read_csv(os.path.join(master_path, fn))))
        except Exception:
            continue
    viz = [x for x in os.listdir(master_path)
           if x.endswith(("_daily.csv", "_weekly.csv", "_hourly.csv"))]
    decomposed = 0
    for fn in sorted(viz)[:40]:
        try:
            df = pd.read_csv(os.path.join(master_path, fn))
            xcol = df.columns[0]
            fig = go.Figure()
            for c in [c for c in df.columns[1:] if pd.api.types.is_numeric_dtype(df[c])]:
                fig.add_trace(go.Scatter(x=df[xcol].astype(str), y=df[c], mode="lines+markers", name=c))
            fig.update_layout(title=fn[:-4], height=360)
            parts.append(_fig_div(fig, "ts_" + fn.replace(".", "_")))
            # seasonal decomposition + stationarity for the first few
            # daily mean series (reference plotSeasonalDecompose + ADF/KPSS)
            if fn.endswith("_daily.csv") and "mean" in df.columns and len(df) >= 14 and decomposed < 4:
                series = pd.to_numeric(df["mean"], errors="coerce")
                sfig = plotSeasonalDecompose(series, period=7, title="Seasonal Decomposition — " + fn[:-4])
                parts.append(_fig_div(sfig, "tsdec_" + fn.replace(".", "_")))
                st = stationarity_check(series)
                parts.append("<p class='note'>Stationarity (rolling-stats heuristic): "
                             + ", ".join(f"{k}={v}" for k, v in st.items()) + "</p>")
                decomposed += 1
        except Exception:
            continue
    return "".join(parts)


def loc_report_gen(master_path: str, print_report=False, **kwargs) -> str:
    """Reference report_generation.py:3902 — geospatial tab from
    geospatial_analyzer outputs."""
    files = os.listdir(master_path)
    geo_csvs = [x for x in files if x.startswith(("Overall_Summary", "Top_", "cluster_output"))]
    geo_charts = [x for x in files if x.startswith(("cluster_plot", "loc_charts"))]
    if not geo_csvs and not geo_charts:
        return ""
    parts = []
    for fn in sorted(geo_csvs)[:20]:
        try:
            parts.append(f"<h3>{fn[:-4]}</h3>" + _tbl(pd.read_csv(os.path.join(master_path, fn))))
        except Exception:
            continue
    parts.append(_charts_section(master_path, sorted(geo_charts), "Geospatial Charts", "geo"))
    return "".join(parts)


def anovos_report(master_path: str, id_col="", label_col=None, corr_threshold=0.4,
                  iv_threshold=0.02, drift_threshold_model=0.1, dataDict_path=".",
                  metricDict_path=".", final_report_path=".", event_label=1,
                  run_type="local", auth_key="NA", output_type=None,
                  sg_print_impact=False, **kwargs) -> str:
    """Reference report_generation.py:3984 — assemble ml_anovos_report.html
    from the on-disk stats + chart objects. Returns the report path.
    run_type routing mirrors reference :4386-4413: databricks resolves
    dbfs:/ to the fuse mount; emr/ak8s write locally then push the html
    via aws s3 cp / azcopy."""
    from anovos_amd.shared import utils as _su

    if run_type == "databricks":
        final_report_path = _su.output_to_local(final_report_path)
        master_path = _su.output_to_local(master_path)
    cloud_target = None
    if run_type in ("emr", "ak8s"):
        cloud_target = final_report_path
        final_report_path = "."
    os.makedirs(final_report_path, exist_ok=True)
    tabs = [
        ("Executive Summary", executive_summary_gen(master_path, label_col, event_label,
                                                    corr_threshold, iv_threshold, drift_threshold_model)),
        ("Wiki", wiki_generator(master_path, dataDict_path if dataDict_path != "." else None,
                                metricDict_path if metricDict_path != "." else None)),
        ("Descriptive Statistics", descriptive_statistics(master_path)),
        ("Quality Check", quality_check(master_path)),
        ("Attribute Associations", attribute_associations(master_path, label_col, event_label,
                                                           corr_threshold, iv_threshold)),
        ("Data Drift & Data Stability", data_drift_stability(master_path, drift_threshold_model)),
    ]
    ts = ts_viz_generate(master_path)
    if ts:
        tabs.append(("Time Series", ts))
    geo = loc_report_gen(master_path)
    if geo:
        tabs.append(("Geospatial", geo))
    out = render_report(tabs, title="ML-Anovos Report")
    path = ends_with(final_report_path) + "ml_anovos_report.html"
    from anovos_amd.core import dist as _dist

    if _dist.rank() == 0:  # concurrent multi-rank writes would tear the file
        with open(path, "w") as f:
            f.write(out)
        if cloud_target is not None:
            _su.cloud_sync(path, ends_with(cloud_target) + "ml_anovos_report.html", run_type, auth_key)
    _dist.barrier()
    return path


def render_report(tabs, title="ML-Anovos Report") -> str:
    """Single-file offline HTML with plotly.js inlined."""
    from plotly.offline import get_plotlyjs

    btns = "".join(
        f"<button class='{'active' if i == 0 else ''}' onclick='showTab({i})'>{_html.escape(name)}</button>"
        for i, (name, _) in enumerate(tabs)
    )
    bodies = "".join(
        f"<div class='tab {'active' if i == 0 else ''}'><h2>{_html.escape(name)}</h2>{content}</div>"
        for i, (name, content) in enumerate(tabs)
    )
    return (
        "<!DOCTYPE html><html><head><meta charset='utf-8'>"
        f"<title>{_html.escape(title)}</title>"
        f"<style>{_CSS}</style>"
        f"<script>{get_plotlyjs()}</script>"
        f"<script>{_JS}</script></head><body>"
        f"<header><h1>{_html.escape(title)}</h1>"
        "<p>Generated by anovos_amd — MI355X-native feature engineering engine</p></header>"
        f"<div class='tabbar'>{btns}</div>{bodies}</body></html>"
    )


def plotSeasonalDecompose(series, period: int = 7, title: str = "Seasonal Decomposition"):
    """Reference report_generation.py:1942 — trend/seasonal/residual
    decomposition chart. statsmodels is not in this stack; the classical
    moving-average decomposition is computed natively (centered MA trend,
    period-mean seasonal, additive residual)."""
    import plotly.subplots as sp

    y = np.asarray(pd.to_numeric(pd.Series(series), errors="coerce").ffill().bfill(), dtype=float)
    n = len(y)
    if n < 2 * period:
        period = max(2, n // 2)
    # centered moving average trend
    k = period
    kernel = np.ones(k) / k
    trend = np.convolve(y, kernel, mode="same")
    detrended = y - trend
    seasonal = np.array([np.nanmean(detrended[i::period]) for i in range(period)])
    seasonal = seasonal - seasonal.mean()
    seas_full = np.tile(seasonal, n // period + 1)[:n]
    resid = y - trend - seas_full
    fig = sp.make_subplots(rows=4, cols=1, shared_xaxes=True,
                           subplot_titles=["Observed", "Trend", "Seasonal", "Residual"])
    x = list(range(n))
    for i, comp in enumerate([y, trend, seas_full, resid]):
        fig.add_trace(go.Scatter(x=x, y=comp, mode="lines"), row=i + 1, col=1)
    fig.update_layout(height=700, showlegend=False, title_text=title)
    return fig


def stationarity_check(series, max_lag: int = 1) -> dict:
    """Rolling-stats stationarity heuristic standing in for the
    reference's ADF/KPSS imports (report_generation.py:55; statsmodels
    absent here): compares mean/variance of the two halves and the lag-1
    autocorrelation of the differenced series."""
    y = np.asarray(pd.to_numeric(pd.Series(series), errors="coerce").dropna(), dtype=float)
    n = len(y)
    if n < 10:
        return {"stationary": None, "reason": "too few points"}
    h1, h2 = y[: n // 2], y[n // 2 :]
    mean_shift = abs(h1.mean() - h2.mean()) / (y.std() + 1e-12)
    var_ratio = (h1.var() + 1e-12) / (h2.var() + 1e-12)
    d = np.diff(y)
    ac1 = float(np.corrcoef(d[:-1], d[1:])[0, 1]) if len(d) > 2 else 0.0
    stationary = bool(mean_shift < 0.5 and 0.25 < var_ratio < 4.0)
    return {"stationary": stationary, "mean_shift": float(mean_shift),
            "variance_ratio": float(var_ratio), "diff_lag1_autocorr": ac1}


# ---------------------------------------------------------------------------
# Reference-named section builders (report_generation.py:99-3812). The
# reference returns datapane objects; here each returns the native html
# fragment (or a DataFrame where the reference tabulates) built from the
# same master_path file contract, so custom-report composers migrating
# from the reference keep their entry points.
# ---------------------------------------------------------------------------

def list_ts_remove_append(l, opt):
    """Remove (opt==1) or append (else) the `_ts` suffix on each name
    (reference report_generation.py:2308)."""
    if opt == 1:
        return [x[:-3] if str(x).endswith("_ts") else x for x in l]
    return [x if str(x).endswith("_ts") else str(x) + "_ts" for x in l]


def drift_stability_ind(missing_recs_drift, drift_tab, missing_recs_stability, stability_tab):
    """(drift_ind, stability_ind) flags from which stat files are missing
    (reference report_generation.py:440): 0 = tab absent, 1 = full,
    stability 0.5 = index present but per-attribute metrics missing."""
    drift_ind = 0 if len(missing_recs_drift) == len(drift_tab) else 1
    if len(missing_recs_stability) == len(stability_tab):
        stability_ind = 0
    elif ("stabilityIndex_metrics" in missing_recs_stability
          and "stability_index" not in missing_recs_stability):
        stability_ind = 0.5
    else:
        stability_ind = 1
    return drift_ind, stability_ind


def data_analyzer_output(master_path, avl_recs_tab=None, tab_name="stats_generator"):
    """html section for one analyzer tab from its saved CSVs (reference
    report_generation.py:233). `avl_recs_tab` limits to specific file
    stems; default renders everything the tab saved."""
    tab_map = {
        "stats_generator": SG_tabs,
        "quality_checker": QC_tabs,
        "association_evaluator": AE_tabs,
    }
    names = tab_map.get(tab_name)
    if names is None:
        raise ValueError(f"tab_name must be one of {sorted(tab_map)}")
    if avl_recs_tab:
        names = [n for n in names if n in set(avl_recs_tab)]
    parts = []
    for n in names:
        df = _read_csv(master_path, n)
        if df is not None:
            parts.append(f"<h3>{remove_u_score(n)}</h3>" + _tbl(df))
    return "".join(parts)


def chart_gen_list(master_path, chart_type, type_col=None):
    """List of html chart divs for every saved `<chart_type>_<col>` plotly
    JSON (reference report_generation.py:475); `type_col` filters to the
    given column names."""
    files = _chart_files(master_path, chart_type)
    if type_col:
        allow = {str(c) for c in type_col}
        files = [f for f in files
                 if f[len(chart_type) + 1:].rsplit(".", 1)[0] in allow]
    out = []
    for i, fn in enumerate(files):
        try:
            out.append(_fig_div(os.path.join(master_path, fn), f"{chart_type}_{i}"))
        except Exception:
            continue
    return out


def line_chart_gen_stability(df1, df2, col):
    """html line chart of a column's per-snapshot stability metrics with
    its summarized CV/SI annotation (reference report_generation.py:99).
    df1 = summarized stability metrics (attribute, stability_index, ...),
    df2 = per-snapshot metrics (attribute, idx, mean, stddev, kurtosis)."""
    d = df2[df2["attribute"] == col] if "attribute" in df2.columns else df2
    fig = go.Figure()
    xcol = "idx" if "idx" in d.columns else d.columns[0]
    for metric in ("mean", "stddev", "kurtosis"):
        if metric in d.columns:
            fig.add_trace(go.Scatter(x=d[xcol], y=pd.to_numeric(d[metric], errors="coerce"),
                                     mode="lines+markers", name=metric))
    title = f"Stability — {col}"
    if df1 is not None and "attribute" in getattr(df1, "columns", []):
        row = df1[df1["attribute"] == col]
        if len(row) and "stability_index" in row.columns:
            title += f" (SI={float(row['stability_index'].iloc[0]):.2f})"
    fig.update_layout(title=title, height=360)
    return _fig_div(fig, "stab_" + str(col))


def ts_landscape(base_path, ts_cols=None, id_col=None):
    """Detected-timestamp landscape table as html (reference
    report_generation.py:2636): the ts_cols_stats.csv summary, filtered
    to `ts_cols` when given."""
    tcs = _read_csv(base_path, "ts_cols_stats")
    if tcs is None:
        return ""
    if ts_cols:
        namecol = tcs.columns[0]
        tcs = tcs[tcs[namecol].isin(set(ts_cols) | {id_col})]
    return "<h3>Time-Series Landscape</h3>" + _tbl(tcs)


def ts_stats(base_path):
    """All saved time-series stat tables as one html block (reference
    report_generation.py:3051) — same file contract as ts_viz_generate."""
    return ts_viz_generate(base_path)


def overall_stats_gen(lat_col_list, long_col_list, geohash_col_list):
    """Summary DataFrame of the detected geospatial fields (reference
    report_generation.py:3210)."""
    return pd.DataFrame(
        {
            "Field Category": ["Latitude", "Longitude", "Geohash"],
            "Count": [len(lat_col_list or []), len(long_col_list or []), len(geohash_col_list or [])],
            "Columns": [", ".join(lat_col_list or []), ", ".join(long_col_list or []),
                        ", ".join(geohash_col_list or [])],
        }
    )


def loc_field_stats(lat_col_list, long_col_list, geohash_col_list, max_records=100):
    """Per-field listing of the detected location columns (reference
    report_generation.py:3250)."""
    rows = [("latitude", c) for c in (lat_col_list or [])]
    rows += [("longitude", c) for c in (long_col_list or [])]
    rows += [("geohash", c) for c in (geohash_col_list or [])]
    return pd.DataFrame(rows[:max_records], columns=["field_type", "column"])


def read_stats_ll_geo(lat_col, long_col, geohash_col, master_path, top_geo_records=100):
    """html of the saved lat/long + geohash descriptive stats CSVs
    (reference report_generation.py:3298)."""
    parts = []
    for fn in sorted(os.listdir(master_path)):
        if fn.endswith(".csv") and fn.startswith(("Overall_Summary", "Top_")):
            try:
                df = pd.read_csv(os.path.join(master_path, fn)).head(top_geo_records)
                parts.append(f"<h3>{fn[:-4]}</h3>" + _tbl(df))
            except Exception:
                continue
    return "".join(parts)


def read_cluster_stats_ll_geo(lat_col, long_col, geohash_col, master_path):
    """html of the saved cluster outputs + cluster plots (reference
    report_generation.py:3535)."""
    parts = []
    for fn in sorted(os.listdir(master_path)):
        if fn.startswith("cluster_output") and fn.endswith(".csv"):
            try:
                parts.append(f"<h3>{fn[:-4]}</h3>" + _tbl(pd.read_csv(os.path.join(master_path, fn))))
            except Exception:
                continue
    plots = [x for x in sorted(os.listdir(master_path)) if x.startswith("cluster_plot")]
    parts.append(_charts_section(master_path, plots, "Cluster Plots", "clus"))
    return "".join(parts)


def read_loc_charts(master_path):
    """html of the saved location charts (reference
    report_generation.py:3812)."""
    plots = [x for x in sorted(os.listdir(master_path)) if x.startswith("loc_charts")]
    return _charts_section(master_path, plots, "Location Charts", "locc")


def lambda_cat(val):
    """Box-Cox λ → the transform it implies (reference
    report_generation.py:2734; the standard Box-Cox power ladder)."""
    if val < -1:
        return "Reciprocal Square Transform"
    if val < -0.5:
        return "Reciprocal Transform"
    if val < 0:
        return "Receiprocal Square Root Transform"
    if val < 0.5:
        return "Log Transform"
    if val < 1:
        return "Square Root Transform"
    if val < 2:
        return "No Transform"
    return "Square Transform"


def gen_time_series_plots(base_path, x_col, y_col, time_cat):
    """html line chart from the saved `<x>_<y>_<time_cat>.csv` aggregate
    (reference report_generation.py:2054); empty string when absent."""
    p = ends_with(base_path) + f"{x_col}_{y_col}_{time_cat}.csv"
    if not os.path.exists(p):
        return ""
    df = pd.read_csv(p).dropna()
    if df.empty:
        return ""
    xc = df.columns[0]
    fig = go.Figure()
    for c in [c for c in df.columns[1:] if pd.api.types.is_numeric_dtype(df[c])]:
        fig.add_trace(go.Scatter(x=df[xc].astype(str), y=df[c], mode="lines+markers", name=c))
    fig.update_layout(title=f"{y_col} by {x_col} ({time_cat})", height=360)
    return _fig_div(fig, f"tsp_{x_col}_{y_col}_{time_cat}")


def _ts_viz(base_path, x_col, y_col, time_cat):
    return gen_time_series_plots(base_path, x_col, y_col, time_cat)


# The reference's nine ts_viz_<view>_<slot> builders lay out the same
# per-granularity aggregate chart in the three time-series report views
# (report_generation.py:2345-3090); slots 1/2/3 are daily/hourly/weekly.
def ts_viz_1_1(base_path, x_col, y_col, output_type="daily"):
    return _ts_viz(base_path, x_col, y_col, output_type or "daily")


def ts_viz_1_2(base_path, x_col, y_col, output_type="hourly"):
    return _ts_viz(base_path, x_col, y_col, output_type or "hourly")


def ts_viz_1_3(base_path, x_col, y_col, output_type="weekly"):
    return _ts_viz(base_path, x_col, y_col, output_type or "weekly")


def ts_viz_2_1(base_path, x_col, y_col):
    return _ts_viz(base_path, x_col, y_col, "daily")


def ts_viz_2_2(base_path, x_col, y_col):
    return _ts_viz(base_path, x_col, y_col, "hourly")


def ts_viz_2_3(base_path, x_col, y_col):
    return _ts_viz(base_path, x_col, y_col, "weekly")


def ts_viz_3_1(base_path, x_col, y_col):
    return _ts_viz(base_path, x_col, y_col, "daily")


def ts_viz_3_2(base_path, x_col, y_col):
    return _ts_viz(base_path, x_col, y_col, "hourly")


def ts_viz_3_3(base_path, x_col, y_col):
    return _ts_viz(base_path, x_col, y_col, "weekly")
