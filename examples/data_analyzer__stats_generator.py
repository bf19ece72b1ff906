#!/usr/bin/env python3
"""Descriptive statistics (reference notebook
examples/notebooks/data_analyzer__stats_generator.ipynb): every
stats_generator metric on a demo frame — fused single-pass kernels on
MI355X, same tidy [attribute, metrics...] outputs as the reference."""

from _common import demo_ctx_and_frame

from anovos_amd.data_analyzer import stats_generator as sg

ctx, idf = demo_ctx_and_frame()
for fn in (sg.global_summary, sg.measures_of_counts, sg.measures_of_centralTendency,
           sg.measures_of_cardinality, sg.measures_of_dispersion,
           sg.measures_of_percentiles, sg.measures_of_shape):
    print(f"--- {fn.__name__}")
    print(fn(ctx, idf, drop_cols=["id"]).to_string(index=False))
