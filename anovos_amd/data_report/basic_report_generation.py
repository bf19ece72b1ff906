"""One-call mini report (reference parity:
``anovos/data_report/basic_report_generation.py`` :95-566 — runs the
stats-generator and quality-checker functions directly on the frame and
saves ``basic_report.html``).

Uses the same native HTML renderer as report_generation (datapane is not
in this stack); all statistics come from the engine's fused GPU kernels.
"""

from __future__ import annotations

import os


from anovos_amd.data_analyzer import association_evaluator as ae
from anovos_amd.data_analyzer import quality_checker as qc
from anovos_amd.data_analyzer import stats_generator as sg
from anovos_amd.data_report.report_generation import _tbl, render_report
from anovos_amd.shared.utils import ends_with


def remove_u_score(col: str) -> str:
    """`_`-separated name -> title-cased words (reference
    basic_report_generation.py:236)."""
    return " ".join(w.title() for w in str(col).split("_"))


_remove_u_score = remove_u_score


def stats_args(path, func):
    """kwargs that wire pre-saved analyzer CSVs into a quality-checker
    call (reference basic_report_generation.py:55-92): stats_unique /
    stats_mode / stats_missing map to the saved cardinality /
    centralTendency / counts CSVs under `path` so detectors reuse the
    already-computed statistics instead of re-scanning the frame."""
    mainfunc_to_args = {
        "biasedness_detection": ["stats_mode"],
        "IDness_detection": ["stats_unique"],
        "nullColumns_detection": ["stats_unique", "stats_mode", "stats_missing"],
        "variable_clustering": ["stats_mode"],
    }
    args_to_statsfunc = {
        "stats_unique": "measures_of_cardinality",
        "stats_mode": "measures_of_centralTendency",
        "stats_missing": "measures_of_counts",
    }
    out = {}
    for arg in mainfunc_to_args.get(func, []):
        out[arg] = {
            "file_path": ends_with(path) + args_to_statsfunc[arg] + ".csv",
            "file_type": "csv",
            "file_configs": {"header": True, "inferSchema": True},
        }
    return out


def anovos_basic_report(ctx, idf, id_col="", label_col="", event_label="",
                        skip_corr_matrix=True, output_path=".", run_type="local",
                        auth_key="NA", print_impact=False, mlflow_config=None) -> str:
    """Reference basic_report_generation.py:95 — returns the report path."""
    local_path = output_path if run_type == "local" else "report_stats"
    os.makedirs(local_path, exist_ok=True)

    SG_funcs = [sg.global_summary, sg.measures_of_counts, sg.measures_of_centralTendency,
                sg.measures_of_cardinality, sg.measures_of_dispersion,
                sg.measures_of_percentiles, sg.measures_of_shape]
    QC_rows_funcs = [qc.duplicate_detection, qc.nullRows_detection]
    QC_cols_funcs = [qc.nullColumns_detection, qc.outlier_detection, qc.IDness_detection,
                     qc.biasedness_detection, qc.invalidEntries_detection]
    AA_funcs = [ae.variable_clustering] if skip_corr_matrix else [ae.correlation_matrix, ae.variable_clustering]
    AT_funcs = [ae.IV_calculation, ae.IG_calculation]

    drop = [c for c in [id_col, label_col] if c]
    sg_parts = []
    for f in SG_funcs:
        try:
            stats = f(ctx, idf, drop_cols=drop)
            # persist like the reference (:167-210): later checkers reuse
            # these via the stats_args wiring instead of re-scanning
            from anovos_amd.core import dist as _dist

            if _dist.rank() == 0:
                stats.to_csv(ends_with(local_path) + f.__name__ + ".csv", index=False)
            _dist.barrier()
            if print_impact:
                print(f.__name__, "\n", stats.to_string(index=False))
            sg_parts.append(f"<h3>{_remove_u_score(f.__name__)}</h3>" + _tbl(stats))
        except Exception as e:
            sg_parts.append(f"<h3>{_remove_u_score(f.__name__)}</h3><p class='note'>failed: {e}</p>")
    qc_parts = []
    for f in QC_rows_funcs + QC_cols_funcs:
        try:
            extra = stats_args(local_path, f.__name__)
            out = f(ctx, idf, drop_cols=drop, **extra) if f not in QC_rows_funcs else f(ctx, idf)
            stats = out[1] if isinstance(out, tuple) else out
            if print_impact:
                print(f.__name__, "\n", stats.to_string(index=False))
            qc_parts.append(f"<h3>{_remove_u_score(f.__name__)}</h3>" + _tbl(stats))
        except Exception as e:
            qc_parts.append(f"<h3>{_remove_u_score(f.__name__)}</h3><p class='note'>failed: {e}</p>")
    aa_parts = []
    for f in AA_funcs + AT_funcs:
        try:
            kwargs = dict(stats_args(local_path, f.__name__))
            if f in AT_funcs:
                if not label_col:
                    continue
                kwargs.update({"label_col": label_col, "event_label": event_label})
            stats = f(ctx, idf, drop_cols=drop, **kwargs)
            aa_parts.append(f"<h3>{_remove_u_score(f.__name__)}</h3>" + _tbl(stats))
        except Exception as e:
            aa_parts.append(f"<h3>{_remove_u_score(f.__name__)}</h3><p class='note'>failed: {e}</p>")

    tabs = [
        ("Descriptive Statistics", "".join(sg_parts)),
        ("Quality Check", "".join(qc_parts)),
        ("Attribute Associations", "".join(aa_parts)),
    ]
    out = render_report(tabs, title="Anovos Basic Report")
    path = ends_with(local_path) + "basic_report.html"
    from anovos_amd.core import dist as _dist

    if _dist.rank() == 0:
        with open(path, "w") as f:
            f.write(out)
    _dist.barrier()
    return path
