#!/usr/bin/env python3
"""Decompose IV/IG section cost at the bench shard shape."""

import sys
import time

import torch

sys.path.insert(0, ".")
from bench import make_synthetic_frame
from anovos_amd.shared.context import init_context


def t(fn, name, n=3):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        r = fn()
    torch.cuda.synchronize()
    print(f"{name}: {(time.perf_counter() - t0) / n * 1000:.2f} ms")
    return r


rows = int(sys.argv[1]) if len(sys.argv) > 1 else 125_000_000
ctx = init_context("cuda")
idf = make_synthetic_frame(rows, ctx.device, seed=5)

from anovos_amd.data_analyzer import association_evaluator as ae
from anovos_amd.data_transformer.transformers import compute_bin_cutoffs, _event_indicator
from anovos_amd.ops import backend
from anovos_amd.shared.utils import attributeType_segregation

num_cols, cat_cols, _ = attributeType_segregation(idf)
num_cols = [c for c in num_cols]
ext = backend.hip_ext()

# prime moments/quantile caches the way the analyzer does
from anovos_amd.ops import histogram as hist_ops, stats as stats_ops

t(lambda: stats_ops.frame_moments(idf, num_cols), "frame_moments (first)", 1)
t(lambda: hist_ops.approx_quantiles(idf, num_cols, [0.01, 0.05, 0.10, 0.25, 0.50, 0.75, 0.90, 0.95, 0.99]),
  "analyzer quantiles (first)", 1)

kept, cuts = t(lambda: compute_bin_cutoffs(ctx, idf, num_cols, "equal_frequency", 10),
               "compute_bin_cutoffs eq_freq (cached ctx)", 1)
label = _event_indicator(idf, "label", "yes")
lab_u8 = label.to(torch.uint8).contiguous()
cut_t = [torch.tensor(c, dtype=torch.float64) for c in cuts]
tensors = [idf.col(c).data.contiguous() for c in kept]
t(lambda: ext.bucketize_label_counts(tensors, cut_t, lab_u8, [12] * len(kept)),
  "bucketize_label_counts kernel (150 num)")
ctensors = [idf.col(c).data.contiguous() for c in cat_cols]
csizes = [len(idf.col(c).dictionary or []) + 1 for c in cat_cols]
t(lambda: ext.label_counts_multi(ctensors, lab_u8, csizes), "label_counts_multi (50 cat)")

idf.clear_stats_cache()
t(lambda: stats_ops.frame_moments(idf, num_cols), "frame_moments (re)", 1)
t(lambda: hist_ops.approx_quantiles(idf, num_cols, [0.01, 0.05, 0.10, 0.25, 0.50, 0.75, 0.90, 0.95, 0.99]),
  "analyzer quantiles (re)", 1)
t(lambda: ae.IV_calculation(ctx, idf, label_col="label", event_label="yes"), "IV full (fresh deciles)", 1)
t(lambda: ae.IG_calculation(ctx, idf, label_col="label", event_label="yes"), "IG full (cached)", 1)
