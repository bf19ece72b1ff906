"""Dataset reader/writer + basic ETL — parity with reference
data_ingest/data_ingest.py (533 LoC).

read/write go through pyarrow into the HBM column store (core/io.py);
unions and joins are tensor concatenations / device hash joins instead
of Spark shuffles.
"""

from __future__ import annotations

import warnings
from typing import Dict, List

import pandas as pd
import torch

from anovos_amd.core import io as core_io
from anovos_amd.core.frame import AnovosFrame, Column
from anovos_amd.shared.utils import attributeType_segregation, pairwise_reduce


def read_dataset(ctx, file_path: str, file_type: str, file_configs: Dict = {}, sharded: bool = True) -> AnovosFrame:
    """Reference data_ingest.py:23-51 — csv/parquet/avro/json.
    sharded=False replicates all parts to every rank (stats artifacts)."""
    device = getattr(ctx, "device", "cpu")
    return core_io.read_dataset(file_path, file_type, file_configs, device=device, sharded=sharded)


def write_dataset(idf: AnovosFrame, file_path: str, file_type: str, file_configs: Dict = {}, column_order: List[str] = []):
    """Reference data_ingest.py:54-117. repartition/coalesce hints are
    accepted and ignored (each rank writes one part: partitioning follows
    the GPU sharding, which is the engine's unit of parallelism) — but a
    negative repartition is still rejected like Spark's
    IllegalArgumentException (reference test_data_ingest_unit.py:79)."""
    rp = (file_configs or {}).get("repartition")
    if rp is not None and int(rp) < 0:
        raise ValueError(f"Invalid repartition value: {rp}")
    core_io.write_dataset(idf, file_path, file_type, file_configs, column_order or None)


def concatenate_dataset(*idfs: AnovosFrame, method_type: str = "name") -> AnovosFrame:
    """Union frames (reference data_ingest.py:120-152): 'name' aligns to
    the first frame's columns, 'index' is positional."""
    if method_type not in ["index", "name"]:
        raise TypeError("Invalid input for concatenate_dataset method")

    def union2(a: AnovosFrame, b: AnovosFrame) -> AnovosFrame:
        if method_type == "name":
            missing = [c for c in a.columns if c not in b.columns]
            if missing:
                raise ValueError(f"columns {missing} missing from a concatenated dataframe")
            b2 = b.select(a.columns)
        else:
            if len(a.columns) != len(b.columns):
                raise ValueError("union by index requires equal column counts")
            b2 = b.rename({bc: ac for ac, bc in zip(a.columns, b.columns)}).select(a.columns)
        cols = {}
        for name in a.columns:
            ca, cb = a.col(name), b2.col(name)
            cols[name] = _concat_columns(ca, cb)
        return AnovosFrame(cols, a.device)

    return pairwise_reduce(union2, idfs)


def _concat_columns(ca: Column, cb: Column) -> Column:
    from anovos_amd.core.dtypes import NULL_CODE

    if ca.kind == "categorical" or cb.kind == "categorical":
        da = ca.dictionary or []
        db = cb.dictionary or []
        union = list(da)
        pos = {s: i for i, s in enumerate(union)}
        for s in db:
            if s not in pos:
                pos[s] = len(union)
                union.append(s)
        lut = torch.tensor([pos[s] for s in db] + [NULL_CODE], dtype=torch.int32, device=cb.data.device)
        codes_b = cb.data.to(torch.long)
        codes_b = torch.where(codes_b == NULL_CODE, torch.full_like(codes_b, len(db)), codes_b)
        newb = lut[codes_b]
        return Column(ca.name, ca.dtype, torch.cat([ca.data, newb.to(ca.data.dtype)]), union)
    dtype = ca.data.dtype if ca.data.dtype.itemsize >= cb.data.dtype.itemsize else cb.data.dtype
    return Column(ca.name, ca.dtype, torch.cat([ca.data.to(dtype), cb.data.to(dtype)]))


def join_dataset(*idfs: AnovosFrame, join_cols, join_type: str) -> AnovosFrame:
    """Device hash join (reference data_ingest.py:155-198). Supported
    join_type: inner, left, full, right, left_semi, left_anti. The right
    frame of each pairwise join is treated as a (broadcastable) lookup —
    in the reference workflow joins attach small auxiliary datasets."""
    if isinstance(join_cols, str):
        join_cols = [x.strip() for x in join_cols.split("|")]
    list_of_df_cols = [x.columns for x in idfs]
    all_cols = [c for sub in list_of_df_cols for c in sub]
    nonjoin = [c for c in all_cols if c not in join_cols]
    if len(nonjoin) != (len(all_cols) - len(list_of_df_cols) * len(join_cols)):
        raise ValueError("Specified join_cols do not match all the Input Dataframe(s)")
    if len(nonjoin) != len(set(nonjoin)):
        raise ValueError("Duplicate column(s) present in non joining column(s) in Input Dataframe(s)")
    return pairwise_reduce(lambda a, b: _join2(a, b, join_cols, join_type), idfs)


def _replicate(b: AnovosFrame) -> AnovosFrame:
    """Gather a (small, row-sharded) side frame onto every rank — the
    engine's broadcast join: the left frame stays sharded, the right
    frame becomes globally visible (Spark shuffles instead; side tables
    in this workload are MBs)."""
    from anovos_amd.core import dist

    if not dist.is_dist():
        return b
    import pandas as pd

    parts = dist.all_gather_object(b.to_pandas())
    full = pd.concat(parts, ignore_index=True)
    return AnovosFrame.from_pandas(full, device=b.device)


def _join2(a: AnovosFrame, b: AnovosFrame, join_cols: List[str], how: str, _replicated: bool = False) -> AnovosFrame:
    from anovos_amd.ops.groupby import row_hash

    if not _replicated:
        b = _replicate(b)
    ha = row_hash(a, join_cols)
    hb = row_hash(b, join_cols).to(ha.device)
    sb, order_b = torch.sort(hb)
    pos = torch.searchsorted(sb, ha)
    pos_c = pos.clamp(max=max(sb.numel() - 1, 0))
    matched = (sb.numel() > 0) & (sb[pos_c] == ha) if sb.numel() else torch.zeros_like(ha, dtype=torch.bool)
    b_idx = order_b[pos_c]

    if how in ("inner", "left_semi"):
        keep_a = matched.nonzero(as_tuple=True)[0]
    elif how == "left_anti":
        keep_a = (~matched).nonzero(as_tuple=True)[0]
    elif how in ("left", "full"):
        keep_a = torch.arange(ha.numel(), device=ha.device)
    elif how == "right":
        # all b rows survive. Matched b rows pair with this rank's a
        # shard; unmatched-vs-ALL-ranks b rows are emitted once, by
        # their owner rank (b is replicated in dist mode).
        inner = _join2(a, b, join_cols, "inner", _replicated=True)
        extra = _unmatched_b_rows(a, b, ha, hb)
        if extra is None or extra.local_rows() == 0:
            return inner.select(a.columns + [c for c in b.columns if c not in join_cols])
        cols = {}
        for name in inner.columns:
            if name in extra.columns:
                cols[name] = extra.col(name)
            else:
                cols[name] = _all_null_column(inner.col(name), extra.local_rows())
        filled = AnovosFrame(cols, inner.device).select(inner.columns)
        return concatenate_dataset(inner, filled, method_type="name")
    else:
        raise ValueError(f"join_type {how} not supported")

    out = a.filter_rows(keep_a)
    if how in ("left_semi", "left_anti"):
        return out
    bsel = b_idx[keep_a]
    bmatched = matched[keep_a]
    for name in b.columns:
        if name in join_cols:
            continue
        cb = b.col(name)
        g = cb.gather(bsel)
        if how in ("left", "full"):
            g.data = _null_where(g, ~bmatched)
        out = out.with_column(name, g)
    if how == "full":
        # append b rows unmatched against the GLOBAL a (ownership-split
        # across ranks so replicated-b extras are emitted exactly once)
        extra = _unmatched_b_rows(a, b, ha, hb)
        if extra is not None and extra.local_rows():
            cols = {}
            for name in out.columns:
                if name in extra.columns:
                    cols[name] = extra.col(name)
                else:
                    src = out.col(name)
                    cols[name] = _all_null_column(src, extra.local_rows())
            out = concatenate_dataset(out, AnovosFrame(cols, out.device).select(out.columns), method_type="name")
    return out


def _unmatched_b_rows(a: AnovosFrame, b: AnovosFrame, ha: torch.Tensor, hb: torch.Tensor):
    """b rows whose key matches NO a row on ANY rank, split by owner
    rank (b identical on all ranks in dist mode)."""
    from anovos_amd.core import dist

    if hb.numel() == 0:
        return None
    bm = torch.zeros(hb.numel(), dtype=torch.bool, device=hb.device)
    if ha.numel():
        sa, _ = torch.sort(ha)
        posb = torch.searchsorted(sa, hb).clamp(max=max(sa.numel() - 1, 0))
        bm = sa[posb] == hb
    if dist.is_dist():
        bm_f = bm.to(torch.float64)
        dist.all_reduce_(bm_f, "max")
        bm = bm_f > 0
        owner = (torch.arange(hb.numel(), device=hb.device) % dist.world_size()) == dist.rank()
        extra_idx = (~bm & owner).nonzero(as_tuple=True)[0]
    else:
        extra_idx = (~bm).nonzero(as_tuple=True)[0]
    if extra_idx.numel() == 0:
        return None
    return b.filter_rows(extra_idx)


def _null_where(col: Column, mask: torch.Tensor) -> torch.Tensor:
    from anovos_amd.core.dtypes import NULL_CODE, NULL_TS

    if col.kind == "numerical":
        return torch.where(mask, torch.full_like(col.data, float("nan")), col.data)
    if col.kind == "categorical":
        return torch.where(mask, torch.full_like(col.data, NULL_CODE), col.data)
    return torch.where(mask, torch.full_like(col.data, NULL_TS), col.data)


def _all_null_column(template: Column, n: int) -> Column:
    from anovos_amd.core.dtypes import NULL_CODE, NULL_TS

    if template.kind == "numerical":
        data = torch.full((n,), float("nan"), dtype=template.data.dtype, device=template.data.device)
    elif template.kind == "categorical":
        data = torch.full((n,), NULL_CODE, dtype=template.data.dtype, device=template.data.device)
    else:
        data = torch.full((n,), NULL_TS, dtype=template.data.dtype, device=template.data.device)
    return Column(template.name, template.dtype, data, template.dictionary)


def delete_column(idf: AnovosFrame, list_of_cols, print_impact=False) -> AnovosFrame:
    """Reference data_ingest.py:201-237."""
    if isinstance(list_of_cols, str):
        list_of_cols = [x.strip() for x in list_of_cols.split("|")]
    odf = idf.drop(list_of_cols)
    if print_impact:
        print("Before: \nNo. of Columns-", len(idf.columns), "\n", idf.columns)
        print("After: \nNo. of Columns-", len(odf.columns), "\n", odf.columns)
    return odf


def select_column(idf: AnovosFrame, list_of_cols, print_impact=False) -> AnovosFrame:
    """Reference data_ingest.py:239-276."""
    if isinstance(list_of_cols, str):
        list_of_cols = [x.strip() for x in list_of_cols.split("|")]
    odf = idf.select(list_of_cols)
    if print_impact:
        print("Before: \nNo. of Columns-", len(idf.columns), "\n", idf.columns)
        print("After: \nNo. of Columns-", len(odf.columns), "\n", odf.columns)
    return odf


def rename_column(idf: AnovosFrame, list_of_cols, list_of_newcols, print_impact=False) -> AnovosFrame:
    """Reference data_ingest.py:277-320."""
    if isinstance(list_of_cols, str):
        list_of_cols = [x.strip() for x in list_of_cols.split("|")]
    if isinstance(list_of_newcols, str):
        list_of_newcols = [x.strip() for x in list_of_newcols.split("|")]
    mapping = dict(zip(list_of_cols, list_of_newcols))
    odf = idf.rename(mapping)
    if print_impact:
        print("Before: \nNo. of Columns-", len(idf.columns), "\n", idf.columns)
        print("After: \nNo. of Columns-", len(odf.columns), "\n", odf.columns)
    return odf


def recast_column(idf: AnovosFrame, list_of_cols, list_of_dtypes, print_impact=False) -> AnovosFrame:
    """Reference data_ingest.py:322-367."""
    if isinstance(list_of_cols, str):
        list_of_cols = [x.strip() for x in list_of_cols.split("|")]
    if isinstance(list_of_dtypes, str):
        list_of_dtypes = [x.strip() for x in list_of_dtypes.split("|")]
    odf = idf
    for c, dt in zip(list_of_cols, list_of_dtypes):
        odf = odf.cast(c, dt.lower())
    if print_impact:
        print("Before: \n", idf.dtypes)
        print("After: \n", odf.dtypes)
    return odf


def recommend_type(ctx, idf, list_of_cols="all", drop_cols=[], dynamic_threshold=0.01, static_threshold=100):
    """Cardinality-based cat/num recommendation — reference
    data_ingest.py:370-533. Returns [attribute, original_form,
    original_dataType, recommended_form, recommended_dataType,
    distinct_value_count]."""
    from anovos_amd.ops import distinct as distinct_ops
    from anovos_amd.shared.utils import normalize_columns

    cols = normalize_columns(idf, list_of_cols, drop_cols)
    schema_cols = [
        "attribute",
        "original_form",
        "original_dataType",
        "recommended_form",
        "recommended_dataType",
        "distinct_value_count",
    ]
    if len(cols) == 0:
        warnings.warn("No recommend_attributeType analysis - No column(s) to analyze")
        return pd.DataFrame(columns=schema_cols)
    if type(dynamic_threshold) != float:
        raise TypeError("Invalid input for dynamic_threshold: float type only")
    if dynamic_threshold <= 0 or dynamic_threshold > 1:
        raise TypeError("Invalid input for dynamic_threshold: Value need to be between 0 and 1")
    if type(static_threshold) != int:
        raise TypeError("Invalid input for static_threshold: int type only")

    from anovos_amd.ops import stats as stats_ops

    num_cols, cat_cols, _ = attributeType_segregation(idf.select(cols))
    distinct = distinct_ops.exact_distinct(idf, num_cols + cat_cols)
    nulls, total = stats_ops.null_counts(idf, num_cols + cat_cols) if (num_cols + cat_cols) else ({}, 0)
    dtype_map = dict(idf.dtypes)
    rows = []
    for col in num_cols:
        fill = total - nulls[col]
        if distinct[col] < min(dynamic_threshold * fill, static_threshold):
            rows.append([col, "numerical", dtype_map[col], "categorical", "string", distinct[col]])
    for col in cat_cols:
        # castable to double with no information loss?
        d = idf.col(col).dictionary or []
        parsed = []
        ok = True
        for s in d:
            try:
                parsed.append(float(s))
            except (TypeError, ValueError):
                ok = False
                break
        if ok and len(set(parsed)) == len(parsed) and len(parsed) > 0:
            fill = total - nulls[col]
            if distinct[col] >= min(dynamic_threshold * fill, static_threshold):
                rows.append([col, "categorical", dtype_map[col], "numerical", "double", distinct[col]])
    if not rows:
        warnings.warn("No column type change recommendation is made")
        return pd.DataFrame(columns=schema_cols)
    return pd.DataFrame(rows, columns=schema_cols)
