#!/usr/bin/env python3
"""Attribute associations (reference notebook
data_analyzer__association_evaluator.ipynb): correlation matrix (bf16
MFMA Gram on MI355X), IV/IG vs a binary label, variable clustering."""

from _common import demo_ctx_and_frame

from anovos_amd.data_analyzer import association_evaluator as ae

ctx, idf = demo_ctx_and_frame()
print(ae.correlation_matrix(ctx, idf, drop_cols=["id", "churn"]).to_string(index=False))
print(ae.IV_calculation(ctx, idf, drop_cols=["id"], label_col="churn",
                        event_label="yes").to_string(index=False))
print(ae.IG_calculation(ctx, idf, drop_cols=["id"], label_col="churn",
                        event_label="yes").to_string(index=False))
print(ae.variable_clustering(ctx, idf, drop_cols=["id", "churn"]).to_string(index=False))
