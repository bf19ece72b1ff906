#!/usr/bin/env python3
"""Datetime functions (reference notebook data_transformer__datetime.ipynb):
parse, extract units, arithmetic, calendar predicates, window aggs."""

import numpy as np
import pandas as pd

from _common import AnovosFrame, init_context

from anovos_amd.data_transformer import datetime as dtf

rng = np.random.default_rng(3)
n = 2000
pdf = pd.DataFrame({
    "id": np.arange(n, dtype=float),
    "ts": (pd.Timestamp("2023-01-01")
           + pd.to_timedelta(rng.integers(0, 365 * 24 * 3600, n), unit="s")).astype(str),
    "v": rng.normal(0, 1, n),
})
ctx = init_context()
idf = AnovosFrame.from_pandas(pdf, device=getattr(ctx, "device", "cpu"))
idf = dtf.string_to_timestamp(ctx, idf, ["ts"], output_mode="replace")
idf = dtf.timeUnits_extraction(idf, ["ts"], ["year", "month", "dayofweek", "hour"], output_mode="append")
idf = dtf.adding_timeUnits(idf, ["ts"], unit="days", unit_value=30, output_mode="append")
idf = dtf.is_weekend(idf, ["ts"], output_mode="append")
print(idf.columns)
print(idf.to_pandas().head(3).to_string(index=False))
