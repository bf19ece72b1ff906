"""Shared helpers for the example scripts: small synthetic frames so
every example runs standalone in seconds on CPU or an MI355X."""

import os
import sys

import numpy as np
import pandas as pd

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from anovos_amd.core.frame import AnovosFrame  # noqa: E402
from anovos_amd.shared.context import init_context  # noqa: E402


def demo_frame(n=5000, seed=7):
    rng = np.random.default_rng(seed)
    pdf = pd.DataFrame({
        "id": [f"u{i:06d}" for i in range(n)],
        "age": rng.integers(18, 90, n).astype(float),
        "income": rng.lognormal(10, 0.6, n),
        "spend": rng.normal(250, 80, n),
        "segment": rng.choice(["bronze", "silver", "gold"], n, p=[0.6, 0.3, 0.1]),
        "churn": rng.choice(["no", "yes"], n, p=[0.8, 0.2]),
    })
    pdf.loc[rng.choice(n, n // 20, replace=False), "spend"] = np.nan
    return pdf


def demo_ctx_and_frame(n=5000):
    ctx = init_context()
    return ctx, AnovosFrame.from_pandas(demo_frame(n), device=getattr(ctx, "device", "cpu"))
