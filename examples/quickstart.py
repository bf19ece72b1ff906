#!/usr/bin/env python3
"""Quickstart: the engine's Python API end-to-end on the synthetic income
dataset (the YAML workflow wraps exactly these calls — config/configs.yaml).

    python tools/make_income_data.py --out data/income_dataset
    python examples/quickstart.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from anovos_amd.data_analyzer import association_evaluator as ae
from anovos_amd.data_analyzer import quality_checker as qc
from anovos_amd.data_analyzer import stats_generator as sg
from anovos_amd.data_ingest.data_ingest import read_dataset
from anovos_amd.data_report import report_preprocessing as rp
from anovos_amd.data_report.report_generation import anovos_report
from anovos_amd.data_transformer import transformers as T
from anovos_amd.shared.context import init_context


def main():
    ctx = init_context()  # local MI355X if visible, else CPU
    print(f"engine context: {ctx}")

    idf = read_dataset(ctx, "data/income_dataset/csv", "csv",
                       {"header": True, "inferSchema": True})
    print(idf)

    # --- descriptive statistics (fused GPU kernels) ---
    print(sg.global_summary(ctx, idf).to_string(index=False))
    print(sg.measures_of_dispersion(ctx, idf, drop_cols=["ifa"]).head().to_string(index=False))

    # --- quality checks with treatment ---
    idf, dup_stats = qc.duplicate_detection(ctx, idf, drop_cols=["ifa"], treatment=True)
    idf, null_stats = qc.nullColumns_detection(ctx, idf, drop_cols=["ifa", "income"],
                                               treatment=True, treatment_method="MMM")
    print(null_stats.head().to_string(index=False))

    # --- associations against the label ---
    iv = ae.IV_calculation(ctx, idf, drop_cols=["ifa"], label_col="income", event_label=">50K")
    print(iv.sort_values("iv", ascending=False).head().to_string(index=False))

    # --- transformations ---
    idf = T.attribute_binning(ctx, idf, ["age", "hours-per-week"],
                              method_type="equal_frequency", bin_size=10, output_mode="append")
    idf = T.z_standardization(ctx, idf, ["capital-gain"], output_mode="append")

    # --- report ---
    rp.save_stats(ctx, sg.global_summary(ctx, idf), "report_stats", "global_summary")
    rp.charts_to_objects(ctx, idf, label_col="income", event_label=">50K",
                         master_path="report_stats", source_path="intermediate_data")
    out = anovos_report(master_path="report_stats", label_col="income",
                        event_label=">50K", final_report_path="report_stats")
    print(f"report: {out}")


if __name__ == "__main__":
    main()
