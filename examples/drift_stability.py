#!/usr/bin/env python3
"""Drift + stability (reference notebook drift_stability.ipynb):
PSI/JSD/HD/KS vs a drifted source, stability index across snapshots."""

import numpy as np
import pandas as pd

from _common import AnovosFrame, init_context

from anovos_amd.drift_stability import drift_detector, stability

rng = np.random.default_rng(17)
n = 20000
ctx = init_context()
dev = getattr(ctx, "device", "cpu")
target = AnovosFrame.from_pandas(pd.DataFrame({"x": rng.normal(0.3, 1.1, n),
                                                "y": rng.lognormal(1, 0.5, n)}), device=dev)
source = AnovosFrame.from_pandas(pd.DataFrame({"x": rng.normal(0.0, 1.0, n),
                                                "y": rng.lognormal(1, 0.5, n)}), device=dev)
d = drift_detector.statistics(ctx, target, idf_source=source, method_type="all", bin_size=10)
print(d.to_string(index=False))
snaps = [AnovosFrame.from_pandas(pd.DataFrame({"m": rng.normal(5 + 0.3 * k, 1, 2000)}), device=dev)
         for k in range(4)]
print(stability.stability_index_computation(ctx, *snaps).to_string(index=False))
