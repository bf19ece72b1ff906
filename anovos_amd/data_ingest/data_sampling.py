"""Random / stratified sampling — parity with reference
data_ingest/data_sampling.py (149 LoC; kernel K17).

Sampling is a counter-based RNG mask on the device (ops/sample.py);
stratified sampling builds the stratum key from the concatenated strata
columns' row hash and derives per-stratum fractions ("population" =
proportionate; "balanced" = equal-count scaled to the smallest stratum,
reference data_sampling.py:140-146)."""

from __future__ import annotations

import warnings

import torch

from anovos_amd.core import dist
from anovos_amd.ops import sample as sample_ops
from anovos_amd.ops.groupby import row_hash


def data_sample(
    idf,
    strata_cols="all",
    drop_cols=[],
    fraction=0.1,
    method_type="random",
    stratified_type="population",
    seed_value=12,
    unique_threshold=0.5,
):
    if type(fraction) != float and type(fraction) != int:
        raise TypeError("Invalid input for fraction")
    if fraction <= 0 or fraction > 1:
        raise TypeError("Invalid input for fraction: fraction value is between 0 and 1")
    if type(seed_value) != int:
        raise TypeError("Invalid input for seed_value")
    if method_type not in ["stratified", "random"]:
        raise TypeError("Invalid input for data_sample method_type")

    if method_type == "random":
        keep = sample_ops.bernoulli_mask(idf.local_rows(), fraction, seed_value, idf.device)
        return idf.filter_rows(keep)

    if type(unique_threshold) != float and type(unique_threshold) != int:
        raise TypeError("Invalid input for unique_threshold")
    if unique_threshold > 1 and type(unique_threshold) != int:
        raise TypeError("Invalid input for unique_threshold: unique_threshold can only be integer if larger than 1")
    if unique_threshold <= 0:
        raise TypeError(
            "Invalid input for unique_threshold: unique_threshold value is either between 0 and 1, or an integer > 1"
        )
    if stratified_type not in ["population", "balanced"]:
        raise TypeError("Invalid input for stratified_type")
    if strata_cols == "all":
        strata_cols = idf.columns
    if isinstance(strata_cols, str):
        strata_cols = [x.strip() for x in strata_cols.split("|")]
    if isinstance(drop_cols, str):
        drop_cols = [x.strip() for x in drop_cols.split("|")]
    strata_cols = [e for e in dict.fromkeys(strata_cols) if e not in drop_cols]
    if len(strata_cols) == 0:
        raise TypeError("Missing strata_cols value")
    from anovos_amd.ops import distinct as distinct_ops

    skip_cols = []
    total = idf.count()
    for col in strata_cols:
        if col not in idf.columns:
            raise TypeError("Invalid input for strata_cols: " + col + " does not exist")
        d = distinct_ops.exact_distinct(idf, [col])[col]
        limit = unique_threshold * total if unique_threshold <= 1 else unique_threshold
        if float(d) > float(limit):
            skip_cols.append(col)
    if skip_cols:
        warnings.warn("Columns dropped from strata due to high cardinality: " + ",".join(skip_cols))
    strata_cols = [e for e in strata_cols if e not in skip_cols]
    if len(strata_cols) == 0:
        warnings.warn("No Stratified Sampling Computation - No strata column(s) to sample")
        return idf

    # strata key: drop rows with null in any strata col, then 64-bit hash
    null_any = torch.zeros(idf.local_rows(), dtype=torch.bool, device=idf.device)
    for c in strata_cols:
        null_any |= idf.col(c).null_mask()
    base = idf.filter_rows(~null_any)
    key = row_hash(base, strata_cols)

    if stratified_type == "population":
        keep = sample_ops.bernoulli_mask(base.local_rows(), fraction, seed_value, base.device)
        return base.filter_rows(keep)

    # balanced: per-stratum fraction scaled so every stratum yields
    # ~ fraction * smallest_stratum rows
    uniq, inv = torch.unique(key, return_inverse=True)
    local_counts = torch.bincount(inv, minlength=uniq.numel()).to(torch.float64)
    if dist.is_dist():
        av = torch.cat(dist.all_gather_tensor(uniq))
        ac = torch.cat(dist.all_gather_tensor(local_counts))
        guniq, ginv = torch.unique(av, return_inverse=True)
        gcounts = torch.zeros(guniq.numel(), dtype=torch.float64, device=av.device)
        gcounts.index_add_(0, ginv, ac)
        pos = torch.searchsorted(guniq, uniq)
        counts = gcounts[pos]
        smallest = float(gcounts.min())  # global smallest stratum, even if absent locally
    else:
        counts = local_counts
        smallest = float(counts.min())
    frac_per = (fraction * smallest / counts).clamp(max=1.0)
    g = torch.Generator(device=base.device)
    g.manual_seed(int(seed_value) * 1000003 + dist.rank())
    u = torch.rand(base.local_rows(), generator=g, device=base.device)
    keep = u < frac_per[inv]
    return base.filter_rows(keep)
