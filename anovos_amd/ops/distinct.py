"""Distinct counts — exact and HyperLogLog (kernel K4).

Reference: uniqueCount_computation (stats_generator.py:529-612) uses
countDistinct or approx_count_distinct(rsd=0.05). Here: categorical
columns get exact distinct from dictionary counts (free); numeric columns
get exact sort-based unique, or HLL (p=12 -> rsd ~1.6%, well inside the
reference's rsd=0.05 default) whose 4K registers merge across ranks with
an all-reduce(max); the fused moments+HLL kernel computes both
prior-free analyzer passes in one frame read.
"""

from __future__ import annotations

import math
from typing import Dict, List

import torch

from anovos_amd.core import dist
from anovos_amd.ops import backend
from anovos_amd.ops.groupby import _mix64

# p=12 -> rsd 1.04/sqrt(4096) ~ 1.6%, well inside the reference default
# rsd=0.05 (stats_generator.py:605-612); 16 KB LDS per block keeps the
# fused kernel at full occupancy (p=14's 64 KB capped it at 2 blocks/CU)
HLL_P = 12
HLL_M = 1 << HLL_P


def hll_registers(t: torch.Tensor, p: int = None) -> torch.Tensor:
    """Local HLL registers (uint8 as int16 tensor [2^p]) for one numeric
    column; NaN skipped. float32 hashes its raw 32 bits through the
    murmur3 finalizer (bit-identical to the HIP kernel's int32 fast
    path); float64 hashes its 8 bytes via splitmix64."""
    p = HLL_P if p is None else p
    m = 1 << p
    if t.is_cuda and backend.use_hip(t):
        ext = backend.hip_ext()
        return ext.hll_registers(t.contiguous(), p)
    v = t[~torch.isnan(t)]
    if v.dtype == torch.float32:
        x = v.view(torch.int32).to(torch.int64) & 0xFFFFFFFF
        h = _fmix32(x)
        idx = h >> (32 - p)
        rem = (h << p) & 0xFFFFFFFF
        rho = _clz32(rem) + 1
        rho = torch.clamp(rho, max=32 - p + 1)
    else:
        x = v.to(torch.float64).view(torch.int64)
        h = _mix64(x)
        idx = (h >> (64 - p)) & (m - 1)
        rem = h << p  # wrapping
        rho = _clz64(rem) + 1
        rho = torch.clamp(rho, max=64 - p + 1)
    regs = torch.zeros(m, dtype=torch.int64, device=t.device)
    regs.scatter_reduce_(0, idx, rho, reduce="amax")
    return regs.to(torch.int16)


_M32 = 0xFFFFFFFF


def _fmix32(x: torch.Tensor) -> torch.Tensor:
    """murmur3 32-bit finalizer on int64 tensors holding uint32 values
    (int64 mul wraps; masking keeps the low 32 bits, which is exactly
    the uint32 product)."""
    x = (x ^ (x >> 16)) & _M32
    x = (x * 0x85EBCA6B) & _M32
    x = (x ^ (x >> 13)) & _M32
    x = (x * 0xC2B2AE35) & _M32
    x = (x ^ (x >> 16)) & _M32
    return x


def _clz32(x: torch.Tensor) -> torch.Tensor:
    """Count leading zeros of the low 32 bits (values in [0, 2^32))."""
    clz = 31 - torch.floor(torch.log2(x.to(torch.float64) + 0.5)).to(torch.int64)
    return torch.where(x == 0, torch.full_like(clz, 32), clz.clamp(0, 31))


def _clz64(x: torch.Tensor) -> torch.Tensor:
    """Count leading zeros of int64 viewed as uint64 (vectorized)."""
    hi = (x >> 32) & 0xFFFFFFFF
    lo = x & 0xFFFFFFFF
    clz_hi = 31 - torch.floor(torch.log2(hi.to(torch.float64) + 0.5)).to(torch.int64)
    clz_hi = torch.where(hi == 0, torch.full_like(clz_hi, 32), clz_hi.clamp(0, 31))
    clz_lo = 31 - torch.floor(torch.log2(lo.to(torch.float64) + 0.5)).to(torch.int64)
    clz_lo = torch.where(lo == 0, torch.full_like(clz_lo, 32), clz_lo.clamp(0, 31))
    return torch.where(hi != 0, clz_hi, 32 + clz_lo)


def hll_estimate(regs: torch.Tensor) -> float:
    """Standard bias-corrected HLL estimate from merged registers."""
    m = float(HLL_M)
    r = regs.to(torch.float64)
    z = torch.pow(2.0, -r).sum().item()
    alpha = 0.7213 / (1 + 1.079 / m)
    e = alpha * m * m / z
    zeros = int((regs == 0).sum())
    if e <= 2.5 * m and zeros:
        e = m * math.log(m / zeros)
    return e


def approx_distinct(idf, cols: List[str]) -> Dict[str, int]:
    """HLL distinct per numeric column; one fused multi-column kernel on
    GPU, registers merged via a single all-reduce(max)."""
    out = {}
    if not cols:
        return out
    first = idf.col(cols[0]).data
    if first.is_cuda and backend.use_hip(first):
        ext = backend.hip_ext()
        cold = [c for c in cols if "moments" not in idf.col(c).cache]
        if len(cold) >= max(2, len(cols) // 2):
            # fused K1/K2+K4: the same read fills the moment cache every
            # later analyzer section will hit (ops/hip moments_hll)
            from anovos_amd.ops import stats as stats_ops

            tensors = [idf.col(c).data.contiguous() for c in cols]
            shifts = stats_ops.compute_column_shifts(tensors)
            mom_local, flat = ext.moments_hll(tensors, HLL_P, shifts)
            glob = stats_ops.merge_moments_global(mom_local).numpy().tolist()
            for i, c in enumerate(cols):
                idf.col(c).cache.setdefault("moments", stats_ops.MomentStats(glob[i], shift=shifts[i]))
        else:
            flat = ext.hll_registers_multi([idf.col(c).data.contiguous() for c in cols], HLL_P)
    else:
        flat = torch.stack([hll_registers(idf.col(c).data) for c in cols]).to(torch.int32)
    dist.all_reduce_(flat, "max")
    regs = flat.cpu().numpy()
    # vectorized estimate across all columns at once
    import numpy as np

    m = float(HLL_M)
    z = np.power(2.0, -regs.astype(np.float64)).sum(axis=1)
    alpha = 0.7213 / (1 + 1.079 / m)
    e = alpha * m * m / z
    zeros = (regs == 0).sum(axis=1)
    small = (e <= 2.5 * m) & (zeros > 0)
    with np.errstate(divide="ignore"):
        lin = m * np.log(np.where(zeros > 0, m / np.maximum(zeros, 1), 1.0))
    e = np.where(small, lin, e)
    # standard 32-bit large-range correction for columns hashed with the
    # 32-bit path (f32 values): above ~2^32/30 the raw estimate
    # undercounts because of hash-space saturation; E* = -2^32 ln(1-E/2^32)
    # (the undercount exceeded the rsd=0.05 contract ~3e8+)
    two32 = 2.0 ** 32
    is32 = np.array([idf.col(c).data.dtype == torch.float32 and idf.col(c).kind != "categorical"
                     for c in cols])
    large = is32 & (e > two32 / 30.0)
    if large.any():
        e = np.where(large, -two32 * np.log(np.maximum(1.0 - e / two32, 1e-15)), e)
    for i, c in enumerate(cols):
        out[c] = int(round(float(e[i])))
    return out


def exact_distinct(idf, cols: List[str]) -> Dict[str, int]:
    """Exact distinct counts (null excluded, as countDistinct does)."""
    from anovos_amd.ops.groupby import cat_value_counts, numeric_value_counts

    out = {}
    cat_cols = [c for c in cols if idf.col(c).kind == "categorical"]
    counts = cat_value_counts(idf, cat_cols) if cat_cols else {}
    for c in cols:
        col = idf.col(c)
        if col.kind == "categorical":
            out[c] = int((counts[c] > 0).sum())
        else:
            vals, _ = numeric_value_counts(idf, c)
            out[c] = int(vals.numel())
    return out
