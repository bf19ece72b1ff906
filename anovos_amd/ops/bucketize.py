"""Bucketize / bin-apply (kernel K6): map numeric values to bin indices
given per-column cutoffs — the apply half of attribute_binning
(reference transformers.py:248-280) and the drift histogram path
(drift_detector.py:216-237).

GPU kernel: all columns in one launch, cutoffs staged in LDS, branchless
binary search per value. Torch path: torch.bucketize per column.
"""

from __future__ import annotations

from typing import List, Sequence

import torch

from anovos_amd.ops import backend


def bucketize_columns(
    tensors: Sequence[torch.Tensor], cutoffs: Sequence[torch.Tensor]
) -> List[torch.Tensor]:
    """Per column: bin index int32 in [0, len(cutoffs_i)] for valid values
    (value <= cutoffs[k] -> bin k semantics, right-closed like Spark
    Bucketizer with +inf upper); null (NaN) -> -1."""
    out = []
    if tensors and tensors[0].is_cuda and backend.use_hip(tensors[0]):
        ext = backend.hip_ext()
        return ext.bucketize_columns(
            [t.contiguous() for t in tensors],
            [c.to(torch.float64).to(tensors[0].device).contiguous() for c in cutoffs],
        )
    for t, cuts in zip(tensors, cutoffs):
        c = cuts.to(torch.float64).to(t.device)
        idx = torch.bucketize(t.to(torch.float64), c, right=False).to(torch.int32)
        idx = torch.where(torch.isnan(t), torch.full_like(idx, -1), idx)
        out.append(idx)
    return out


def bucketize_columns_float(tensors: Sequence[torch.Tensor], cutoffs: Sequence[torch.Tensor]) -> List[torch.Tensor]:
    """Fused bucketize emitting the binned-column layout directly:
    float32 (bin index + 1), NaN for null — what attribute_binning
    materializes (saves three elementwise passes per column)."""
    if tensors and tensors[0].is_cuda and backend.use_hip(tensors[0]):
        ext = backend.hip_ext()
        return ext.bucketize_columns_float(
            [t.contiguous() for t in tensors],
            [c.to(torch.float64).to(tensors[0].device).contiguous() for c in cutoffs],
        )
    out = []
    for t, cuts in zip(tensors, cutoffs):
        c = cuts.to(torch.float64).to(t.device)
        idx = (torch.bucketize(t.to(torch.float64), c, right=False) + 1).to(torch.float32)
        idx = torch.where(torch.isnan(t), torch.full_like(idx, float("nan")), idx)
        out.append(idx)
    return out
