"""groupBy-count family (kernel K5/K21): category frequencies, mode,
top-k, label-conditioned histograms for IV/IG.

Categorical columns are dictionary codes, so groupBy-count is a bincount
over the (small) dictionary — LDS-staged on GPU — merged across ranks
with one all-reduce per batch of columns. Exact numeric mode/unique use
sort-based torch.unique (rocPRIM radix sort under the hood on ROCm).

Reference semantics: stats_generator.mode_computation (:328, groupBy-
count-top1), outlier_categories ranking (transformers.py:3614-3641),
duplicate detection (quality_checker.py:122).
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch

from anovos_amd.core import dist
from anovos_amd.core.dtypes import NULL_CODE
from anovos_amd.ops import backend


def align_dictionaries(idf, cols: List[str]) -> None:
    """Make categorical dictionaries rank-identical (in place).

    Ingest produces sorted per-rank dictionaries, so low-cardinality
    columns (every shard sees every category) align for free — but
    high-cardinality columns built per rank (e.g. lat/long→geohash
    strings) do not, and every dictionary-indexed collective
    (cat_value_counts all-reduce, IV/IG label counts, drift binning)
    requires identical code→value maps on all ranks. This detects
    misalignment with one signature gather (stable blake2b — Python's
    str hash is salted per process) and heals only the misaligned
    columns: union-sort the gathered dictionaries, remap local codes by
    LUT. Columns are shared between derived frames, so one heal fixes
    every frame holding the column. Collective: must be called on all
    ranks with the same cols."""
    if not dist.is_dist():
        return
    # skip columns already verified/healed (flag set identically on all
    # ranks — same call sequence — so the skip stays rank-uniform and
    # the hot path pays no collective after the first touch)
    cat = [
        c for c in cols
        if idf.col(c).kind == "categorical" and "dict_aligned" not in idf.col(c).cache
    ]
    if not cat:
        return
    import hashlib

    sigs = []
    for c in cat:
        d = idf.col(c).dictionary or []
        h = hashlib.blake2b("\x00".join(map(str, d)).encode(), digest_size=8).hexdigest()
        sigs.append((len(d), h))
    gathered = dist.all_gather_object(sigs)
    bad = [i for i in range(len(cat)) if len({g[i] for g in gathered}) > 1]
    for i in bad:
        col = idf.col(cat[i])
        d = list(col.dictionary or [])
        union = sorted(set().union(*[set(x) for x in dist.all_gather_object(d)]))
        pos = {v: k for k, v in enumerate(union)}
        dev = col.data.device
        lut = torch.tensor([pos[v] for v in d] + [NULL_CODE], dtype=torch.int64, device=dev)
        codes = col.data.long()
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(d)), codes)
        col.data = lut[codes].to(col.data.dtype)
        col.dictionary = union
        col.cache.clear()
    for c in cat:
        idf.col(c).cache["dict_aligned"] = True


def cat_value_counts(idf, cols: List[str]) -> Dict[str, torch.Tensor]:
    """Global counts per dictionary code for categorical columns.
    Returns {col: int64 tensor [dict_size]} on CPU. Cached per Column."""
    align_dictionaries(idf, cols)
    out = {}
    cached = [c for c in cols if "cat_counts" in idf.col(c).cache]
    for c in cached:
        out[c] = idf.col(c).cache["cat_counts"]
    cols = [c for c in cols if c not in set(cached)]
    bufs = []
    gpu_cols = [c for c in cols if idf.col(c).data.is_cuda and backend.use_hip(idf.col(c).data)]
    if gpu_cols:
        # one fused launch for every uncached categorical column (K5);
        # the per-column null slot feeds the null-count cache for free
        ext = backend.hip_ext()
        sizes = [len(idf.col(c).dictionary or []) for c in gpu_cols]
        flat_multi = ext.code_counts_multi([idf.col(c).data.contiguous() for c in gpu_cols], sizes)
        gpu_cnt = {}
        off = 0
        for c, size in zip(gpu_cols, sizes):
            gpu_cnt[c] = flat_multi[off : off + size]  # stays on device
            gpu_cnt[c + "\0null"] = flat_multi[off + size : off + size + 1]
            off += size + 1
    for c in cols:
        col = idf.col(c)
        size = len(col.dictionary or [])
        codes = col.data
        if c in gpu_cols:
            cnt = gpu_cnt[c]
        else:
            valid = codes[codes != NULL_CODE].long()
            cnt = torch.bincount(valid, minlength=size) if size else torch.zeros(0, dtype=torch.int64, device=codes.device)
        bufs.append(cnt)
    if bufs:
        flat = torch.cat([b for b in bufs]) if len(bufs) > 1 else bufs[0]
        dist.all_reduce_(flat, "sum")
        flat_host = flat.cpu()  # ONE sync for all columns
        off = 0
        for c, b in zip(cols, bufs):
            out[c] = flat_host[off : off + b.numel()]
            idf.col(c).cache["cat_counts"] = out[c]
            off += b.numel()
    if gpu_cols:
        null_host = torch.cat([gpu_cnt[c + "\0null"] for c in gpu_cols]).cpu()
        for i, c in enumerate(gpu_cols):
            idf.col(c).cache.setdefault("nulls_local", int(null_host[i]))
    return out


def discrete_modes(idf, cols: List[str]) -> Dict[str, Tuple[Optional[float], int]]:
    """Exact modes for integral numeric columns in ONE fused dense
    histogram launch (no sorts). Columns that aren't integral or span a
    range > 4096 fall back to per-column numeric_value_counts."""
    from anovos_amd.ops import histogram as hist_ops
    from anovos_amd.ops import stats as _stats

    out: Dict[str, Tuple[Optional[float], int]] = {}
    if not cols:
        return out
    moments = _stats.frame_moments(idf, cols)
    dense, rest = [], []
    for c in cols:
        m = moments[c]
        if m.n > 0 and m.integral and m.min == m.min and (m.max - m.min) < 4096:
            dense.append(c)
        else:
            rest.append(c)
    if dense:
        import numpy as np

        # cached dense per-integer counts (filled by the exact integral
        # quantile path — same single frame read serves both)
        cached = [c for c in dense if ("inthist",) in idf.col(c).cache]
        for c in cached:
            h, mn = idf.col(c).cache[("inthist",)]
            idx = int(np.argmax(h))
            cnt = int(h[idx])
            out[c] = (float(mn + idx), cnt) if cnt > 0 else (None, 0)
        dense = [c for c in dense if c not in set(cached)]
    if dense:
        import numpy as np

        R = [int(moments[c].max - moments[c].min) + 1 for c in dense]
        M = max(R)
        tensors = [idf.col(c).data for c in dense]
        lo = torch.tensor([moments[c].min for c in dense], dtype=torch.float64)
        hi = torch.tensor([moments[c].min + r for c, r in zip(dense, R)], dtype=torch.float64)
        hist = hist_ops.global_histograms(tensors, lo, hi, M).cpu().numpy()
        for i, c in enumerate(dense):
            idx = int(np.argmax(hist[i]))
            cnt = int(hist[i][idx])
            val = moments[c].min + int(idx * R[i] / M)
            out[c] = (float(val), cnt) if cnt > 0 else (None, 0)
    for c in rest:
        m = moments[c]
        if dist.is_dist() and m.n > 1_000_000 and not m.integral:
            # EXACT continuous mode at scale (replaces the r1 histogram
            # approximation): local unique+count, then hash-partitioned
            # all-to-all so each unique value crosses the fabric exactly
            # once (not an all-gather of ~n values to every rank)
            t = idf.col(c).data
            x = t[~torch.isnan(t)]
            vals, cnts = torch.unique(x, return_counts=True)
            out[c] = _global_mode_partitioned(vals, cnts)
            continue
        vals, cnts = numeric_value_counts(idf, c)
        if vals.numel() == 0:
            out[c] = (None, 0)
        else:
            i = int(torch.argmax(cnts).item())
            out[c] = (float(vals[i]), int(cnts[i]))
    return out


def _global_mode_partitioned(vals: torch.Tensor, cnts: torch.Tensor):
    """Exact global (mode, count) from per-rank local value counts via a
    hash-partitioned exchange: rank q owns the values whose bit-hash maps
    to q, merges their counts, picks its partition's top-1; the world's
    top-1 is an all-gather of world_size tiny candidates. Deterministic
    tie-break: highest count, then smallest value."""
    ws = dist.world_size()
    v64 = vals.to(torch.float64)
    v64 = torch.where(v64 == 0, torch.zeros_like(v64), v64)  # canonicalize -0.0
    part = _mix64(v64.view(torch.int64)).remainder(ws)
    send_v = [v64[part == q] for q in range(ws)]
    send_c = [cnts.to(torch.int64)[part == q] for q in range(ws)]
    rv = torch.cat([t for t in dist.all_to_all_tensor(send_v)])
    rc = torch.cat([t for t in dist.all_to_all_tensor(send_c)])
    if rv.numel():
        uv, inv = torch.unique(rv, return_inverse=True)
        uc = torch.zeros(uv.numel(), dtype=torch.int64, device=rv.device)
        uc.index_add_(0, inv, rc)
        mx = int(uc.max())
        cand = float(uv[uc == mx].min())
        local_best = (mx, cand)
    else:
        local_best = (0, float("nan"))
    best = max(dist.all_gather_object(local_best), key=lambda t: (t[0], -t[1] if t[1] == t[1] else float("-inf")))
    if best[0] == 0:
        return (None, 0)
    return (best[1], best[0])


def mode(idf, col: str, counts: Optional[torch.Tensor] = None) -> Tuple[Optional[str], int]:
    """Global mode (value, count). Works for categorical (dictionary) and
    numeric (exact unique) columns. Ties: reference takes Spark's
    groupBy().count().orderBy(desc).limit(1) — an arbitrary max; we take
    the first max (lowest code / smallest value)."""
    c = idf.col(col)
    if c.kind == "categorical":
        if counts is None:
            counts = cat_value_counts(idf, [col])[col]
        if counts.numel() == 0 or int(counts.sum()) == 0:
            return None, 0
        i = int(torch.argmax(counts).item())
        return c.dictionary[i], int(counts[i])
    vals, cnts = numeric_value_counts(idf, col)
    if vals.numel() == 0:
        return None, 0
    i = int(torch.argmax(cnts).item())
    return float(vals[i]), int(cnts[i])


def numeric_value_counts(idf, col: str) -> Tuple[torch.Tensor, torch.Tensor]:
    """Exact (value, count) for a numeric column, merged across ranks.

    Fast path: integral-valued columns with bounded range use a dense
    bincount (one pass, no sort) — exact, and the common case for
    discrete numerics (ages, counts, codes). Otherwise sort-based
    torch.unique (rocPRIM radix sort on ROCm)."""
    t = idf.col(col).data
    x = t[~torch.isnan(t)]
    m = idf.col(col).cache.get("moments")
    if m is None:
        from anovos_amd.ops import stats as _stats

        m = _stats.frame_moments(idf, [col])[col]
    if x.numel() and m is not None and m.min == m.min and (m.max - m.min) < 4_000_000:
        if m.integral or (m.n_frac != m.n_frac and bool((x == torch.trunc(x)).all())):
            xi = x.to(torch.int64)
            lo = int(m.min)
            cnts_d = torch.bincount(xi - lo, minlength=int(m.max) - lo + 1)
            nz = cnts_d.nonzero(as_tuple=True)[0]
            vals = (nz + lo).to(t.dtype)
            cnts = cnts_d[nz]
            if dist.is_dist():
                return _merge_value_counts(vals, cnts)
            return vals.cpu(), cnts.cpu()
    vals, cnts = torch.unique(x, return_counts=True)
    if dist.is_dist():
        return _merge_value_counts(vals, cnts)
    return vals.cpu(), cnts.cpu()


def _merge_value_counts(vals: torch.Tensor, cnts: torch.Tensor):
    # tensorized cross-rank merge (device-resident under RCCL): gather
    # both varlen vectors, then unique+index_add on device
    av = torch.cat(dist.all_gather_tensor(vals))
    ac = torch.cat(dist.all_gather_tensor(cnts.to(torch.int64)))
    uv, inv = torch.unique(av, return_inverse=True)
    uc = torch.zeros(uv.numel(), dtype=torch.int64, device=av.device)
    uc.index_add_(0, inv, ac)
    return uv.cpu(), uc.cpu()


def duplicate_row_count(idf, cols: Optional[List[str]] = None) -> int:
    """Rows minus distinct rows (reference quality_checker.py:122:
    groupBy-all-cols). Implemented as a 64-bit row-hash count-distinct —
    the K5 dedup design. Exact up to hash collisions (~n^2/2^64)."""
    cols = cols or idf.columns
    h = row_hash(idf, cols)
    uniq = torch.unique(h)
    if dist.is_dist():
        nuniq = torch.unique(torch.cat(dist.all_gather_tensor(uniq))).numel()
    else:
        nuniq = uniq.numel()
    total = idf.count()
    return int(total - nuniq)


def row_hash(idf, cols: List[str]) -> torch.Tensor:
    """64-bit combined row hash across columns (device-side)."""
    n = idf.local_rows()
    dev = idf.device
    h = torch.full((n,), -0x61C8864680B583EB, dtype=torch.int64, device=dev)  # 0x9E3779B97F4A7C15
    for c in cols:
        col = idf.col(c)
        if col.kind == "numerical":
            b = col.data.to(torch.float64).view(torch.int64)
            b = torch.where(torch.isnan(col.data), torch.full_like(b, -1), b)
        elif col.kind == "categorical":
            # hash the string VALUES (dictionary codes differ across frames):
            # stable 64-bit string hash LUT gathered by code
            lut = torch.tensor(
                [_str_hash64(s) for s in (col.dictionary or [])] + [-1],
                dtype=torch.int64,
                device=dev,
            )
            codes = col.data.to(torch.long)
            codes = torch.where(codes == NULL_CODE, torch.full_like(codes, lut.numel() - 1), codes)
            b = lut[codes]
        else:
            b = col.data.to(torch.int64)
        h = _mix64(h ^ _mix64(b))
    return h


def _str_hash64(s: str) -> int:
    """Stable FNV-1a 64-bit string hash (python ints, wrapped to int64)."""
    h = 0xCBF29CE484222325
    for byte in s.encode("utf-8"):
        h ^= byte
        h = (h * 0x100000001B3) & 0xFFFFFFFFFFFFFFFF
    return h - 0x10000000000000000 if h >= 0x8000000000000000 else h


def _lshr(x: torch.Tensor, n: int) -> torch.Tensor:
    """Logical (zero-fill) right shift on int64 (torch >> is arithmetic)."""
    return (x >> n) & ((1 << (64 - n)) - 1)


def _mix64(x: torch.Tensor) -> torch.Tensor:
    # splitmix64 finalizer (wrapping int64 arithmetic, LOGICAL right shifts
    # to match the uint64 device kernel bit-for-bit)
    x = x ^ _lshr(x, 30)
    x = x * -0x40A7B892E31B1A47  # 0xBF58476D1CE4E5B9
    x = x ^ _lshr(x, 27)
    x = x * -0x6B2FB644ECCEEE15  # 0x94D049BB133111EB
    x = x ^ _lshr(x, 31)
    return x
