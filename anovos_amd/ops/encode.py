"""Dictionary encoding / decoding utilities (kernel K12 support).

- build/apply StringIndexer-style label maps (frequencyDesc order like
  Spark ML StringIndexer, reference transformers.py:652-733),
- global dictionary unification across ranks (all-gather of host dicts +
  device-side code remap via a LUT gather),
- numeric <-> string column conversion.
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch

from anovos_amd.core import dist
from anovos_amd.core.dtypes import NULL_CODE
from anovos_amd.core.frame import Column


def numeric_to_string_column(col: Column) -> Column:
    """Cast a numeric column to a string (categorical) column."""
    t = col.data
    vals = torch.unique(t[~torch.isnan(t)])
    vals_np = vals.cpu().numpy()
    strs = [_fmt_num(v) for v in vals_np]
    # map each value to its unique index
    idx = torch.searchsorted(vals, torch.nan_to_num(t, nan=float("inf")))
    idx = torch.clamp(idx, max=max(vals.numel() - 1, 0)).to(torch.int32)
    idx = torch.where(torch.isnan(t), torch.full_like(idx, NULL_CODE), idx)
    return Column(col.name, "string", idx, strs)


def _fmt_num(v) -> str:
    f = float(v)
    if f == int(f) and abs(f) < 1e15:
        return str(int(f))
    return repr(f)


def unify_dictionaries(cols: List[Column]) -> List[Column]:
    """Make the dictionary of each categorical column identical across
    ranks (required before cross-rank count merges). All-gathers the host
    dictionaries, builds the union in sorted order, remaps device codes
    with one LUT gather per column."""
    if not dist.is_dist():
        return cols
    out = []
    for col in cols:
        local_dict = col.dictionary or []
        gathered = dist.all_gather_object(local_dict)
        union = sorted(set().union(*[set(g) for g in gathered]))
        if union == local_dict:
            out.append(col)
            continue
        pos = {s: i for i, s in enumerate(union)}
        lut = torch.tensor([pos[s] for s in local_dict] + [NULL_CODE], dtype=torch.int32, device=col.data.device)
        codes = col.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(local_dict)), codes)
        out.append(Column(col.name, col.dtype, lut[codes], union))
    return out


def index_map_from_counts(dictionary: List[str], counts: torch.Tensor, order: str = "frequencyDesc") -> Dict[str, int]:
    """StringIndexer label->index map. frequencyDesc: most frequent label
    gets 0; ties broken alphabetically (Spark behavior)."""
    items = list(zip(dictionary, counts.tolist()))
    if order == "frequencyDesc":
        items.sort(key=lambda kv: (-kv[1], kv[0]))
    elif order == "alphabetAsc":
        items.sort(key=lambda kv: kv[0])
    else:
        raise ValueError(order)
    return {k: i for i, (k, _) in enumerate(items)}


def apply_index_maps_batch(cols: List[Column], mappings: List[Dict[str, int]]) -> List[torch.Tensor]:
    """Batched StringIndexer apply: ONE fused LUT-gather kernel covers all
    columns (K12). Unseen/null -> NaN float32 outputs."""
    from anovos_amd.ops import backend

    if cols and cols[0].data.is_cuda and backend.use_hip(cols[0].data):
        ext = backend.hip_ext()
        luts = [
            torch.tensor(
                [float(m.get(s, float("nan"))) for s in (c.dictionary or [])] or [float("nan")],
                dtype=torch.float32,
            )
            for c, m in zip(cols, mappings)
        ]
        return ext.lut_apply_f32([c.data.contiguous() for c in cols], luts)
    return [apply_index_map(c, m) for c, m in zip(cols, mappings)]


def remap_codes_batch(cols: List[Column], luts: List[torch.Tensor]) -> List[torch.Tensor]:
    """Batched int32 code remap (outlier_categories etc.): out = lut[code],
    null -> -1, via one fused kernel."""
    from anovos_amd.ops import backend

    if cols and cols[0].data.is_cuda and backend.use_hip(cols[0].data):
        ext = backend.hip_ext()
        return ext.lut_apply_i32([c.data.contiguous() for c in cols], [l.to(torch.int32) for l in luts])
    out = []
    for c, lut in zip(cols, luts):
        lut_l = torch.cat([lut.to(torch.int64), torch.tensor([NULL_CODE])]).to(c.data.device)
        codes = c.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, lut.numel()), codes)
        out.append(lut_l[codes].to(torch.int32))
    return out


def apply_index_map(col: Column, mapping: Dict[str, int], unseen: int = -1) -> torch.Tensor:
    """Apply a label->index map to a categorical column; returns float32
    tensor with NaN for null/unseen (matches reference's numeric output
    columns)."""
    lut_vals = [float(mapping.get(s, unseen)) for s in (col.dictionary or [])]
    lut = torch.tensor(lut_vals + [float("nan")], dtype=torch.float32, device=col.data.device)
    codes = col.data.to(torch.long)
    codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(lut_vals)), codes)
    vals = lut[codes]
    vals = torch.where(vals == unseen, torch.full_like(vals, float("nan")), vals)
    return vals
