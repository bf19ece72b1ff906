"""Sampling kernels (K17): Bernoulli random masks and stratified
fractions — reference data_sampling.py:127-148 (df.sample / sampleBy).

Counter-based RNG (philox via torch generator seeded per rank) so results
are reproducible per (seed, rank)."""

from __future__ import annotations

from typing import Dict, Optional

import torch

from anovos_amd.core import dist
from anovos_amd.core.dtypes import NULL_CODE


def bernoulli_mask(n: int, fraction: float, seed: int, device) -> torch.Tensor:
    device = torch.device(device)
    g = torch.Generator(device=device)
    g.manual_seed(int(seed) * 1000003 + dist.rank())
    u = torch.rand(n, generator=g, device=device)
    return u < fraction


def stratified_mask(codes: torch.Tensor, fractions: Dict[int, float], seed: int) -> torch.Tensor:
    """Per-stratum Bernoulli mask over an int32 code column."""
    n = codes.shape[0]
    g = torch.Generator(device=codes.device)
    g.manual_seed(int(seed) * 1000003 + dist.rank())
    u = torch.rand(n, generator=g, device=codes.device)
    frac = torch.zeros(int(codes.max().item()) + 2 if n else 1, dtype=torch.float32, device=codes.device)
    for k, f in fractions.items():
        if 0 <= int(k) < frac.numel():
            frac[int(k)] = float(f)
    idx = codes.to(torch.long).clamp(min=0)
    keep = u < frac[idx]
    keep = keep & (codes != NULL_CODE)
    return keep
