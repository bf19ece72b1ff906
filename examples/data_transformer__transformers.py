#!/usr/bin/env python3
"""Transformers (reference notebook data_transformer__transformers.ipynb):
binning, encoding, scaling, imputation, math transforms, box-cox."""

from _common import demo_ctx_and_frame

from anovos_amd.data_transformer import transformers as T

ctx, idf = demo_ctx_and_frame()
odf = T.attribute_binning(ctx, idf, ["age", "income"], method_type="equal_frequency",
                          bin_size=5, output_mode="append")
print([c for c in odf.columns if c.endswith("_binned")])
odf = T.cat_to_num_unsupervised(ctx, odf, ["segment"], method_type=1, output_mode="append")
odf = T.z_standardization(ctx, odf, ["spend"], output_mode="append")
odf = T.imputation_MMM(ctx, odf, ["spend"])
odf = T.feature_transformation(odf, ["income"], method_type="ln", output_mode="append")
odf = T.boxcox_transformation(odf, ["income"], output_mode="append")
print(odf.columns)
