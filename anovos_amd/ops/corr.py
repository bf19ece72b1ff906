"""Correlation / covariance matrix (kernel K8): standardize-then-Gram.

Reference: association_evaluator.correlation_matrix (:118-123) assembles
a vector column and calls MLlib Correlation.corr (one JVM pass). Here the
Pearson matrix is computed from a single fused pass producing the Gram
matrix X^T X of the (mean-centered) column block:

- HIP path: hand-written bf16 MFMA tile kernel with on-the-fly
  standardization (ops/hip/corr_mfma.hip), fp32 accumulate; per-GPU
  partial Gram matrices merged with one RCCL all-reduce.
- Library path (sanctioned): torch.matmul (rocBLAS) on the centered
  block, also the CPU backend.

Null handling matches the reference: correlation_matrix is run on an
imputed frame (workflow imputes MMM first); any remaining NaN is treated
as the column mean (contributes zero to covariance).
"""

from __future__ import annotations

from typing import List

import numpy as np
import torch

from anovos_amd.core import dist
from anovos_amd.ops import backend
from anovos_amd.ops import stats as stats_ops


def pearson_matrix(idf, cols: List[str], moments=None, use_bf16: bool = True) -> np.ndarray:
    """Global Pearson correlation matrix over numeric columns."""
    if moments is None:
        moments = stats_ops.frame_moments(idf, cols)
    dev = idf.device
    n = idf.count()
    tensors = [idf.col(c).data for c in cols]
    col_means = [m if m == m else 0.0 for m in (moments[c].mean for c in cols)]
    if dev.type == "cuda" and backend.use_hip(tensors[0]) and use_bf16:
        ext = backend.hip_ext()
        # f32 columns: the kernel centers in-register (subtraction of
        # nearby f32 values is exact, error = storage quantization).
        # f64 columns MUST center in f64 BEFORE the downcast: casting
        # raw values loses the whole signal when |mean| >> spread
        # (f32 spacing at 1e8 is 8.0).
        f32, kmeans = [], []
        for t, m in zip(tensors, col_means):
            if t.dtype == torch.float32:
                f32.append(t.contiguous())
                kmeans.append(m)
            else:
                f32.append((t - m).to(torch.float32).contiguous())
                kmeans.append(0.0)
        means = torch.tensor(kmeans, dtype=torch.float32, device=dev)
        gram = ext.centered_gram_bf16(f32, means)  # [k,k] fp32
    else:
        # center in f64, then downcast; NaN contributes zero (== mean)
        X = torch.stack(
            [torch.nan_to_num((t.to(torch.float64) - m).to(torch.float32), nan=0.0)
             for t, m in zip(tensors, col_means)],
            dim=1,
        )
        gram = X.T @ X
    gram = gram.to(torch.float64)
    if dist.is_dist():
        dist.all_reduce_(gram, "sum")
    g = gram.cpu().numpy()
    d = np.sqrt(np.clip(np.diag(g), 1e-300, None))
    corr = g / np.outer(d, d)
    np.fill_diagonal(corr, 1.0)
    return np.clip(corr, -1.0, 1.0)


def covariance_matrix(idf, cols: List[str], moments=None) -> np.ndarray:
    """Global covariance matrix (n-1 denominator, Spark RowMatrix
    computeCovariance semantics — reference association_eval_varclus.py:71-84)."""
    if moments is None:
        moments = stats_ops.frame_moments(idf, cols)
    tensors = [idf.col(c).data for c in cols]
    col_means = [m if m == m else 0.0 for m in (moments[c].mean for c in cols)]
    # center in f64 before any precision loss; NaN contributes zero
    Xc = torch.stack(
        [torch.nan_to_num(t.to(torch.float64) - m, nan=0.0) for t, m in zip(tensors, col_means)],
        dim=1,
    )
    gram = Xc.T @ Xc
    if dist.is_dist():
        dist.all_reduce_(gram, "sum")
    n = idf.count()
    return (gram / max(n - 1, 1)).cpu().numpy()
