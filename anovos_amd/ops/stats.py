"""Fused multi-column statistics (kernel K1/K2 of SURVEY.md §2.10).

One pass over all numeric columns produces, per column, the fp64 partial
vector [n_valid, s1, s2, s3, s4, min, max, zero_count, n_frac] where
s1..s4 are power sums about a per-column pivot sampled from the data
(numerically stable skew/kurt at |mean| >> stddev). The
reference computes these with one Spark job per column per statistic
(stats_generator.py:485-494 — the dominant anti-pattern); here every
column is covered by a single kernel launch and ONE batched RCCL
all-reduce merges the 8-double-per-column partials across ranks.

HIP path: ops/hip/anovos_kernels.hip (partials per (col, chunk) block,
deterministic two-kernel reduce, fp64 accumulators).
Torch path: reference implementation, also the CPU backend.
"""

from __future__ import annotations

import math
from typing import Dict, List, Sequence

import torch

from anovos_amd.core import dist
from anovos_amd.ops import backend

NSTAT = 9  # n, s1, s2, s3, s4, min, max, zeros, n_frac (non-integral count)


def compute_column_shifts(tensors: Sequence[torch.Tensor]) -> List[float]:
    """Per-column accumulation pivots for numerically stable moments.

    Raw power sums Σx..Σx⁴ cancel catastrophically when |mean| >> stddev
    (skew/kurt lose ALL digits at mean/sd ≳ 1e5 in fp64 — the SURVEY §6
    hard-part; Spark avoids it with incremental central moments). We
    instead accumulate about a pivot sampled from the data: the median of
    the first 64 non-null values (robust to a leading outlier). Every
    rank must use the SAME pivot so the s1..s4 all-reduce(sum) stays
    valid — an all-reduce(min) over per-rank candidates picks one
    deterministically. All-null columns shift by 0."""
    if len(tensors) == 0:
        return []
    import numpy as np

    heads = torch.cat([t[:64].to(torch.float64) for t in tensors]).cpu().numpy()
    shifts, off = [], 0
    for t in tensors:
        k = min(64, t.numel())
        h = heads[off : off + k]
        off += k
        h = h[~np.isnan(h)]
        shifts.append(float(np.median(h)) if h.size else float("inf"))
    if dist.is_dist():
        dev = tensors[0].device
        tt = torch.tensor(shifts, dtype=torch.float64,
                          device=dev if dev.type == "cuda" else "cpu")
        dist.all_reduce_(tt, "min")
        shifts = tt.cpu().tolist()
    return [0.0 if s == float("inf") else s for s in shifts]


def column_moments_local(tensors: Sequence[torch.Tensor], shifts: Sequence[float] = None) -> torch.Tensor:
    """Local partial moment vectors for a list of numeric columns
    (NaN = null). Returns fp64 tensor [ncols, NSTAT] on the columns'
    device; s1..s4 are power sums of (x - shift) per column."""
    if len(tensors) == 0:
        return torch.empty(0, NSTAT, dtype=torch.float64)
    if shifts is None:
        shifts = [0.0] * len(tensors)
    dev = tensors[0].device
    if dev.type == "cuda" and backend.use_hip(tensors[0]):
        ext = backend.hip_ext()
        return ext.column_moments([t.contiguous() for t in tensors], list(shifts))
    out = torch.empty(len(tensors), NSTAT, dtype=torch.float64, device=dev)
    for i, t in enumerate(tensors):
        td = t.to(torch.float64)
        valid = ~torch.isnan(td)
        x = torch.where(valid, td, torch.zeros_like(td))
        d = torch.where(valid, td - shifts[i], torch.zeros_like(td))
        n = valid.sum()
        s1 = d.sum()
        d2 = d * d
        s2 = d2.sum()
        s3 = (d2 * d).sum()
        s4 = (d2 * d2).sum()
        if int(n) > 0:
            mn = td[valid].min()
            mx = td[valid].max()
        else:
            mn = torch.tensor(float("nan"), dtype=torch.float64, device=dev)
            mx = torch.tensor(float("nan"), dtype=torch.float64, device=dev)
        zeros = ((x == 0) & valid).sum()
        nfrac = ((x != torch.trunc(x)) & valid).sum()
        out[i] = torch.stack([n.to(torch.float64), s1, s2, s3, s4, mn, mx, zeros.to(torch.float64), nfrac.to(torch.float64)])
    return out


def merge_moments_global(local: torch.Tensor) -> torch.Tensor:
    """RCCL merge of [ncols, NSTAT] partials: sum for n/s1..s4/zeros/n_frac, min/max
    for the extrema — batched into two fused all-reduces."""
    if not dist.is_dist():
        return local.cpu()
    sums = local[:, [0, 1, 2, 3, 4, 7, 8]].contiguous()
    mn = torch.nan_to_num(local[:, 5], nan=float("inf")).contiguous()
    mx = torch.nan_to_num(local[:, 6], nan=float("-inf")).contiguous()
    dist.all_reduce_(sums, "sum")
    dist.all_reduce_(mn, "min")
    dist.all_reduce_(mx, "max")
    out = torch.empty_like(local)
    out[:, [0, 1, 2, 3, 4, 7, 8]] = sums
    out[:, 5] = torch.where(torch.isinf(mn), torch.full_like(mn, float("nan")), mn)
    out[:, 6] = torch.where(torch.isinf(mx), torch.full_like(mx, float("nan")), mx)
    return out.cpu()


class MomentStats:
    """Derived statistics for one column from its global moment vector.
    s1..s4 are power sums about `shift` (the accumulation pivot); the
    central-moment conversion below is exact algebra and stays accurate
    because |mean - shift| is O(data spread), not O(|mean|)."""

    __slots__ = ("n", "s1", "s2", "s3", "s4", "min", "max", "zeros", "n_frac", "shift")

    def __init__(self, vec, shift: float = 0.0):
        vals = [float(v) for v in vec]
        if len(vals) == 8:  # legacy 8-slot vector
            vals.append(float("nan"))
        if len(vals) == 10:  # shift appended host-side
            shift = vals.pop()
        (self.n, self.s1, self.s2, self.s3, self.s4, self.min, self.max,
         self.zeros, self.n_frac) = vals
        self.shift = shift

    @property
    def integral(self):
        """True when every valid value is integer-valued (enables dense
        exact mode/unique counting without a sort)."""
        return self.n_frac == 0

    @property
    def mean(self):
        return self.shift + self.s1 / self.n if self.n > 0 else float("nan")

    def _central(self):
        # δ = mean - shift; central moments from shifted power sums
        n = self.n
        d = self.s1 / n
        M2 = self.s2 - n * d * d
        M3 = self.s3 - 3 * d * self.s2 + 2 * n * d**3
        M4 = self.s4 - 4 * d * self.s3 + 6 * d * d * self.s2 - 3 * n * d**4
        return max(M2, 0.0), M3, M4

    @property
    def variance(self):  # sample variance (Spark summary stddev is n-1)
        if self.n < 2:
            return float("nan")
        M2, _, _ = self._central()
        return M2 / (self.n - 1)

    @property
    def stddev(self):
        v = self.variance
        return math.sqrt(v) if v == v else float("nan")

    @property
    def skewness(self):  # Spark F.skewness: population, biased
        if self.n < 1:
            return float("nan")
        M2, M3, _ = self._central()
        if M2 <= 0:
            return float("nan")
        m2 = M2 / self.n
        m3 = M3 / self.n
        return m3 / m2**1.5

    @property
    def kurtosis(self):  # Spark F.kurtosis: excess kurtosis, population
        if self.n < 1:
            return float("nan")
        M2, _, M4 = self._central()
        if M2 <= 0:
            return float("nan")
        m2 = M2 / self.n
        m4 = M4 / self.n
        return m4 / (m2 * m2) - 3.0


def frame_moments(idf, cols: List[str]) -> Dict[str, MomentStats]:
    """Global moment stats for the given numeric columns of a frame.
    Cached per Column (the reference's stats-reuse wiring, workflow.py:91-145);
    benchmarks clear the cache per step to keep timing honest."""
    out: Dict[str, MomentStats] = {}
    todo = []
    for c in cols:
        m = idf.col(c).cache.get("moments")
        if m is not None:
            out[c] = m
        else:
            todo.append(c)
    if todo:
        tensors = [idf.col(c).data for c in todo]
        shifts = compute_column_shifts(tensors)
        local = column_moments_local(tensors, shifts)
        glob = merge_moments_global(local).numpy().tolist()  # one conversion, not 9/col
        for i, c in enumerate(todo):
            m = MomentStats(glob[i], shift=shifts[i])
            idf.col(c).cache["moments"] = m
            out[c] = m
    return out


def null_counts(idf, cols: List[str]) -> Dict[str, int]:
    """Per-column global null counts (any kind). Numeric columns derive
    the count from the fused moments pass (total - n); categorical and
    other columns use one small reduction each, all-reduced in a batch."""
    total = idf.count()
    res: Dict[str, int] = {}
    num_cols = [c for c in cols if idf.col(c).kind == "numerical" and "nulls" not in idf.col(c).cache]
    if num_cols:
        moments = frame_moments(idf, num_cols)
        for c in num_cols:
            idf.col(c).cache["nulls"] = int(total - moments[c].n)
    other = [c for c in cols if "nulls" not in idf.col(c).cache]
    if other:
        dev = idf.device
        # categorical nulls on GPU ride along with the fused code-count
        # launch (K5 null slot, cached as nulls_local by cat_value_counts)
        cat_gpu = [c for c in other if idf.col(c).kind == "categorical"
                   and idf.col(c).data.is_cuda and "nulls_local" not in idf.col(c).cache]
        if cat_gpu:
            from anovos_amd.ops import groupby as groupby_ops

            groupby_ops.cat_value_counts(idf, cat_gpu)
        # assemble host-side (a per-element GPU tensor write is a tiny
        # kernel + sync each); only uncached non-categorical columns need
        # a device reduction
        vals = [None] * len(other)
        pending = []
        for i, c in enumerate(other):
            cached_local = idf.col(c).cache.get("nulls_local")
            if cached_local is not None:
                vals[i] = float(cached_local)
            else:
                pending.append(i)
        if pending:
            sums = torch.stack([idf.col(other[i]).null_mask().sum() for i in pending]).cpu()
            for k, i in enumerate(pending):
                vals[i] = float(sums[k])
        local = torch.tensor(vals, dtype=torch.float64, device=dev if dist.is_dist() else "cpu")
        dist.all_reduce_(local, "sum")
        local_l = local.cpu().numpy().tolist()
        for i, c in enumerate(other):
            idf.col(c).cache["nulls"] = int(local_l[i])
    for c in cols:
        res[c] = idf.col(c).cache["nulls"]
    return res, total
