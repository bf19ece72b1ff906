"""Fused row-wise scans (kernel K10): per-row null count, per-row outlier
flags, row filters over a columnar layout.

Reference: nullRows_detection UDF (quality_checker.py:248-258) counts
nulls per row; the GPU kernel streams every column once, accumulating
per-row counters staged in LDS (rows x 2B), instead of a per-row Python
UDF."""

from __future__ import annotations

from typing import List

import torch

from anovos_amd.ops import backend


def row_null_counts(idf, cols: List[str]) -> torch.Tensor:
    """int32 tensor [local_rows] of null counts across the given columns."""
    n = idf.local_rows()
    dev = idf.device
    first = idf.col(cols[0]).data if cols else None
    if first is not None and first.is_cuda and backend.use_hip(first):
        ext = backend.hip_ext()
        # one fused launch covers numeric (NaN) AND categorical (-1 code)
        # columns; timestamps (INT64_MIN null) use the eager fallback
        fused = [idf.col(c).data.contiguous() for c in cols
                 if idf.col(c).kind == "numerical" or idf.col(c).data.dtype == torch.int32]
        rest = [c for c in cols
                if not (idf.col(c).kind == "numerical" or idf.col(c).data.dtype == torch.int32)]
        out = torch.zeros(n, dtype=torch.int32, device=dev)
        if fused:
            ext.row_null_counts_num(fused, out)
        for c in rest:
            out += idf.col(c).null_mask().to(torch.int32)
        return out
    out = torch.zeros(n, dtype=torch.int32, device=dev)
    for c in cols:
        out += idf.col(c).null_mask().to(torch.int32)
    return out
