#!/usr/bin/env python3
"""Quality checks with treatments (reference notebook
data_analyzer__quality_checker.ipynb): duplicates, null rows/columns,
outliers, IDness, biasedness, invalid entries."""

from _common import demo_ctx_and_frame

from anovos_amd.data_analyzer import quality_checker as qc

ctx, idf = demo_ctx_and_frame()
odf, d = qc.duplicate_detection(ctx, idf, drop_cols=["id"], treatment=True)
print(d.to_string(index=False))
odf, s = qc.nullColumns_detection(ctx, odf, drop_cols=["id", "churn"],
                                  treatment=True, treatment_method="MMM")
print(s.to_string(index=False))
odf, o = qc.outlier_detection(ctx, odf, list_of_cols=["income", "spend"],
                              detection_side="both", treatment=True,
                              treatment_method="value_replacement")
print(o.to_string(index=False))
for fn in (qc.IDness_detection, qc.biasedness_detection, qc.invalidEntries_detection):
    _, st = fn(ctx, odf, drop_cols=["id"])
    print(f"--- {fn.__name__}")
    print(st.to_string(index=False))
