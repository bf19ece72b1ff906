#!/usr/bin/env python3
"""Time-series inspection (reference notebook
data_analyzer__ts_analyzer.ipynb): eligibility stats + per-bucket viz
aggregates over a detected timestamp column."""

import tempfile

import numpy as np
import pandas as pd

from _common import AnovosFrame, init_context

from anovos_amd.data_analyzer import ts_analyzer as tsa
from anovos_amd.data_ingest import ts_auto_detection as tsad

rng = np.random.default_rng(11)
n = 3000
pdf = pd.DataFrame({
    "id": rng.integers(0, 300, n).astype(float),
    "ts": (pd.Timestamp("2023-01-01")
           + pd.to_timedelta(rng.integers(0, 90 * 24 * 3600, n), unit="s")).astype(str),
    "amount": rng.lognormal(3, 0.5, n),
    "channel": rng.choice(["web", "store", "app"], n),
})
ctx = init_context()
idf = AnovosFrame.from_pandas(pdf, device=getattr(ctx, "device", "cpu"))
with tempfile.TemporaryDirectory() as td:
    odf, ts_cols, _, _ = tsad.ts_preprocess(ctx, idf, id_col="id", output_path=td)
    out_cols = tsa.ts_analyzer(ctx, odf, id_col="id", max_days=3600, output_path=td)
    print("analyzed ts columns:", out_cols)
    import os

    print(sorted(os.listdir(td))[:8])
