"""Columnar IO: csv/parquet/json/avro <-> AnovosFrame via pyarrow.

Replaces the Spark readers of the reference (data_ingest.py:23-117). A
"path" may be a file or a directory of part files (Spark writes
directories). Row-partitioning: with world_size W, rank r reads part
files (or row-group ranges) r, r+W, r+2W... so ingest is parallel and no
rows are duplicated; writers emit one part file per rank.

Avro: a minimal pure-python Avro Object Container File codec
(core/avro_codec.py) — the arrow ecosystem here has no avro reader and
the reference needs avro read/write parity (data_ingest.py:36-38).
"""

from __future__ import annotations

import glob
import json
import os
from typing import Dict, List, Optional

import pandas as pd
import pyarrow as pa
import pyarrow.csv as pacsv
import pyarrow.parquet as papq

from anovos_amd.core import dist
from anovos_amd.core.frame import AnovosFrame


def _expand_parts(file_path: str, exts) -> List[str]:
    if os.path.isdir(file_path):
        parts = []
        for e in exts:
            parts += glob.glob(os.path.join(file_path, f"*{e}"))
        parts = sorted(p for p in parts if not os.path.basename(p).startswith(("_", ".")))
        if not parts:
            # Spark part files may have no extension for csv/json
            parts = sorted(
                p
                for p in glob.glob(os.path.join(file_path, "part-*"))
                if not p.endswith(".crc")
            )
        return parts
    return [file_path]


def _my_parts(parts: List[str], sharded: bool = True) -> List[str]:
    if not sharded:
        return parts
    w, r = dist.world_size(), dist.rank()
    if w <= 1:
        return parts
    mine = parts[r::w]
    return mine


def read_dataset(file_path: str, file_type: str, file_configs: Dict = None, device="cpu",
                 sharded: bool = True) -> AnovosFrame:
    """sharded=True (default): part files round-robin across ranks (row
    partitioning). sharded=False: EVERY rank reads all parts — required
    for small stats/model artifacts consumed via the stats-reuse
    contract (a sharded read would hand rank>0 zero parts of a
    single-file stats CSV)."""
    file_configs = dict(file_configs or {})
    ft = file_type.lower()
    if ft == "csv":
        idf = _read_csv(file_path, file_configs, device, sharded)
    elif ft == "parquet":
        idf = _read_parquet(file_path, file_configs, device, sharded)
    elif ft == "json":
        idf = _read_json(file_path, file_configs, device, sharded)
    elif ft == "avro":
        idf = _read_avro(file_path, file_configs, device, sharded)
    else:
        raise ValueError(f"unsupported file_type: {file_type}")
    return _unify_frame_dictionaries(_reconcile_schema(idf))


def _reconcile_schema(idf: AnovosFrame) -> AnovosFrame:
    """Make per-rank inferred schemas identical before any
    schema-shaped collective.

    Sharded reads infer dtypes from each rank's own part files; a
    header-only part (an empty shard written by another rank) infers
    every column as string, and an all-null column can infer numeric on
    one shard and string on another. A rank-dependent dtype changes the
    categorical column set — and with it the number of collectives in
    _unify_frame_dictionaries — which deadlocks. One tiny object gather
    elects the schema of the first rank that holds rows; deviating
    ranks coerce their (usually empty) columns to it."""
    from anovos_amd.core import dist

    if not dist.is_dist():
        return idf
    local = [(n, idf.col(n).dtype) for n in idf.columns]
    gathered = dist.all_gather_object((idf.local_rows(), local))
    schemas = [sch for _, sch in gathered]
    if all(s == schemas[0] for s in schemas):
        return idf
    ref = dict(next((sch for r, sch in gathered if r > 0), schemas[0]))
    out = idf
    for name, dt in local:
        want = ref.get(name, dt)
        if dt != want:
            out = out.with_column(name, _coerce_column(out.col(name), want))
    return out


def _coerce_column(col, want: str):
    """Convert a Column to the elected dtype (schema reconciliation)."""
    import torch

    from anovos_amd.core.frame import Column
    from anovos_amd.core.dtypes import NULL_CODE, NULL_TS, kind_of_dtype, is_timestamp_dtype

    n = int(col.data.numel())
    dev = col.data.device
    want_kind = "other" if is_timestamp_dtype(want) else kind_of_dtype(want)
    if n == 0:
        if want_kind == "numerical":
            return Column(col.name, want, torch.empty(0, dtype=torch.float64, device=dev))
        if is_timestamp_dtype(want):
            return Column(col.name, want, torch.empty(0, dtype=torch.int64, device=dev))
        return Column(col.name, want, torch.empty(0, dtype=torch.int32, device=dev), [])
    if col.kind == "categorical" and want_kind == "numerical":
        d = col.dictionary or []
        def _f(s):
            try:
                return float(s)
            except (TypeError, ValueError):
                return float("nan")
        lut = torch.tensor([_f(s) for s in d] + [float("nan")], dtype=torch.float64, device=dev)
        codes = col.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(d)), codes)
        return Column(col.name, want, lut[codes])
    if col.kind == "numerical" and want_kind == "categorical":
        from anovos_amd.ops.encode import _fmt_num

        x = col.data.to(torch.float64)
        null = torch.isnan(x)
        uniq, inv = torch.unique(torch.nan_to_num(x), return_inverse=True)
        strs = [_fmt_num(float(u)) for u in uniq.cpu()]
        order = sorted(range(len(strs)), key=lambda i: strs[i])
        rankpos = torch.empty(len(strs), dtype=torch.int64, device=dev)
        rankpos[torch.tensor(order, device=dev)] = torch.arange(len(strs), device=dev)
        codes = rankpos[inv].to(torch.int32)
        codes = torch.where(null, torch.full_like(codes, NULL_CODE), codes)
        return Column(col.name, want, codes, [strs[i] for i in order])
    if col.kind == "categorical" and is_timestamp_dtype(want):
        import pandas as _pd

        d = col.dictionary or []
        parsed = _pd.to_datetime(_pd.Series(d), errors="coerce")
        vals = [int(v.value // 1000) if v == v else NULL_TS for v in parsed]
        lut = torch.tensor(vals + [NULL_TS], dtype=torch.int64, device=dev)
        codes = col.data.to(torch.long)
        codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(d)), codes)
        return Column(col.name, want, lut[codes])
    if col.kind == "numerical" and want_kind == "numerical":
        return Column(col.name, want, col.data)
    raise ValueError(
        f"schema reconciliation cannot coerce column {col.name!r} from {col.dtype} to {want}"
    )


def _unify_frame_dictionaries(idf: AnovosFrame) -> AnovosFrame:
    """Multi-rank reads shard rows by file part, so each rank's
    first-seen category set can differ; every cross-rank count merge
    assumes positionally identical dictionaries. One all-gather of the
    (host, tiny) dictionaries at ingest unifies them for the frame's
    lifetime."""
    from anovos_amd.core import dist

    if not dist.is_dist():
        return idf
    from anovos_amd.ops.encode import unify_dictionaries

    cats = [n for n in idf.columns if idf.col(n).kind == "categorical"]
    if not cats:
        return idf
    unified = unify_dictionaries([idf.col(n) for n in cats])
    out = idf
    for col in unified:
        out = out.with_column(col.name, col)
    return out


def _concat_tables(tables: List[pa.Table]) -> pa.Table:
    if not tables:
        return pa.table({})
    return pa.concat_tables(tables, promote_options="permissive")


def _to_frame(table: pa.Table, device) -> AnovosFrame:
    """Arrow table → AnovosFrame WITHOUT a pandas object-array detour:
    strings dictionary-encode in Arrow C++ (then remap to the engine's
    sorted dictionary order), numerics/timestamps convert zero-copy-ish.
    The previous to_pandas() + per-column factorize path dominated
    ingest (~8 s of an 11.5 s 14-dataset read at 1M rows). Columns with
    types this fast path doesn't know fall back to the pandas route,
    keeping dtype semantics identical."""
    import numpy as np
    import torch

    import pyarrow.compute as pc

    from anovos_amd.core.dtypes import NULL_CODE, NULL_TS
    from anovos_amd.core.frame import Column, _column_from_series

    dev = torch.device(device)
    cols = {}
    for name in table.column_names:
        ca = table.column(name)
        arr = ca.combine_chunks() if isinstance(ca, pa.ChunkedArray) else ca
        if isinstance(arr, pa.ChunkedArray):  # zero chunks edge
            arr = arr.combine_chunks()
        t = arr.type
        try:
            if pa.types.is_dictionary(t):
                darr = arr
                dictionary = [("" if v is None else str(v)) for v in darr.dictionary.to_pylist()]
                raw = darr.indices.to_numpy(zero_copy_only=False)
                # null indices surface as float NaN: normalize BEFORE the
                # int cast (NaN->int is platform-defined)
                if raw.dtype.kind == "f":
                    raw = np.nan_to_num(raw, nan=-1.0)
                codes = raw.astype(np.int64, copy=True)
                null = np.asarray(pc.is_null(darr).to_numpy(zero_copy_only=False))
                cols[name] = _sorted_dict_column(name, codes, dictionary, null, dev)
                continue
            if pa.types.is_string(t) or pa.types.is_large_string(t):
                denc = pc.dictionary_encode(arr)
                dictionary = [("" if v is None else str(v)) for v in denc.dictionary.to_pylist()]
                codes = denc.indices.to_numpy(zero_copy_only=False)
                null = np.asarray(pc.is_null(arr).to_numpy(zero_copy_only=False))
                codes = np.where(null, -1, np.nan_to_num(codes.astype(np.float64), nan=-1)).astype(np.int64)
                cols[name] = _sorted_dict_column(name, codes, dictionary, null, dev)
                continue
            if pa.types.is_timestamp(t) or pa.types.is_date(t):
                ts = pc.cast(arr, pa.timestamp("us"))
                vals = ts.to_numpy(zero_copy_only=False).astype("datetime64[us]").astype(np.int64)
                null = np.asarray(pc.is_null(arr).to_numpy(zero_copy_only=False))
                vals = np.where(null, NULL_TS, vals)
                cols[name] = Column(name, "timestamp", torch.from_numpy(np.ascontiguousarray(vals)).to(dev))
                continue
            if pa.types.is_floating(t):
                vals = arr.to_numpy(zero_copy_only=False)
                out = vals.astype(np.float64 if t == pa.float64() else np.float32, copy=False)
                dt = "double" if out.dtype == np.float64 else "float"
                cols[name] = Column(name, dt, torch.from_numpy(
                    np.require(out, requirements=["C", "W"])).to(dev))
                continue
            if pa.types.is_boolean(t):
                vals = arr.to_numpy(zero_copy_only=False).astype(np.float32)
                cols[name] = Column(name, "int", torch.from_numpy(np.ascontiguousarray(vals)).to(dev))
                continue
            if pa.types.is_integer(t):
                if arr.null_count:
                    # pandas would surface these as float64 -> "double"
                    vals = arr.to_numpy(zero_copy_only=False).astype(np.float64)
                    cols[name] = Column(name, "double", torch.from_numpy(np.ascontiguousarray(vals)).to(dev))
                else:
                    wide = t.bit_width > 32
                    vals = arr.to_numpy(zero_copy_only=False).astype(np.float64 if wide else np.float32)
                    cols[name] = Column(name, "bigint" if wide else "int",
                                        torch.from_numpy(np.ascontiguousarray(vals)).to(dev))
                continue
        except Exception:
            pass
        # fallback: pandas semantics for anything else
        s = arr.to_pandas()
        cols[name] = _column_from_series(name, s, dev)
    return AnovosFrame(cols, dev)


def _sorted_dict_column(name, codes_i64, dictionary, null_mask, dev):
    """Remap arrow first-occurrence dictionary codes onto the engine's
    SORTED dictionary order (cross-rank determinism contract)."""
    import numpy as np
    import torch

    from anovos_amd.core.dtypes import NULL_CODE
    from anovos_amd.core.frame import Column

    cats = np.asarray(dictionary, dtype=object)
    order = np.argsort(cats.astype(str))
    remap = np.empty(max(len(cats), 1), dtype=np.int32)
    remap[order] = np.arange(len(cats), dtype=np.int32)
    safe = np.maximum(codes_i64, 0)
    out = np.where((codes_i64 >= 0) & ~null_mask, remap[np.minimum(safe, max(len(cats) - 1, 0))], NULL_CODE).astype(np.int32)
    sorted_dict = [str(c) for c in cats[order]]
    return Column(name, "string", torch.from_numpy(
        np.require(out, requirements=["C", "W"])).to(dev), sorted_dict)


def _empty_like_first(parts_all, read_one):
    """Fewer part files than ranks: this rank holds an EMPTY shard with
    the schema of part 0 (the reference's Spark reader parallelism never
    constrains part count; every merge protocol tolerates 0-row shards)."""
    t = read_one(parts_all[0])
    return t.slice(0, 0)


def _read_csv(path, cfg, device, sharded=True):
    header = str(cfg.get("header", True)).lower() in ("true", "1")
    delim = cfg.get("delimiter", cfg.get("sep", ","))
    parts_all = _expand_parts(path, [".csv"])
    parts = _my_parts(parts_all, sharded)
    tables = []
    for p in parts:
        ro = pacsv.ReadOptions(autogenerate_column_names=not header)
        po = pacsv.ParseOptions(delimiter=delim)
        co = pacsv.ConvertOptions(strings_can_be_null=True)
        if str(cfg.get("inferSchema", True)).lower() not in ("true", "1"):
            pass  # arrow always infers; parity is close enough (ints/floats/strings)
        tables.append(pacsv.read_csv(p, read_options=ro, parse_options=po, convert_options=co))
    if not tables:
        if not parts_all:
            raise FileNotFoundError(f"no csv part files under {path}")
        ro = pacsv.ReadOptions(autogenerate_column_names=not header)
        po = pacsv.ParseOptions(delimiter=delim)
        co = pacsv.ConvertOptions(strings_can_be_null=True)
        tables = [_empty_like_first(parts_all, lambda p: pacsv.read_csv(p, read_options=ro, parse_options=po, convert_options=co))]
    return _to_frame(_concat_tables(tables), device)


def _read_parquet(path, cfg, device, sharded=True):
    parts_all = _expand_parts(path, [".parquet", ".pq"])
    parts = _my_parts(parts_all, sharded)
    if not parts:
        if not parts_all:
            raise FileNotFoundError(f"no parquet files under {path}")
        return _to_frame(_empty_like_first(parts_all, papq.read_table), device)
    tables = [papq.read_table(p) for p in parts]
    return _to_frame(_concat_tables(tables), device)


def _read_json(path, cfg, device, sharded=True):
    import pyarrow.json as pajson

    parts_all = _expand_parts(path, [".json", ".jsonl"])
    parts = _my_parts(parts_all, sharded)
    tables = [pajson.read_json(p) for p in parts]
    if not tables:
        if not parts_all:
            raise FileNotFoundError(f"no json files under {path}")
        tables = [_empty_like_first(parts_all, pajson.read_json)]
    return _to_frame(_parse_iso_timestamp_strings(_concat_tables(tables)), device)


def _parse_iso_timestamp_strings(table: pa.Table) -> pa.Table:
    """Arrow's JSON reader leaves ISO-8601 timestamps as strings (the
    writer emits date_format='iso' for Spark parity); detect and parse
    string columns whose values are ISO timestamps."""
    import re

    import pyarrow.compute as pc

    for i, name in enumerate(table.column_names):
        col = table.column(name)
        arr = col.combine_chunks() if isinstance(col, pa.ChunkedArray) else col
        if not pa.types.is_string(arr.type) or len(arr) == 0:
            continue
        sample = next((v for v in arr.to_pylist()[:50] if v), None)
        if not (sample and re.match(r"^\d{4}-\d{2}-\d{2}T\d{2}:\d{2}:\d{2}", str(sample))):
            continue
        try:
            # arrow's ISO parser handles fractional seconds (strptime's
            # %f does not)
            ts = pc.cast(arr, pa.timestamp("us"))
            if pc.sum(pc.is_null(ts)).as_py() == pc.sum(pc.is_null(arr)).as_py():
                table = table.set_column(i, name, ts)
        except Exception:
            pass
    return table


def _read_avro(path, cfg, device, sharded=True):
    from anovos_amd.core import avro_codec

    parts_all = _expand_parts(path, [".avro"])
    parts = _my_parts(parts_all, sharded)
    if not parts:
        if not parts_all:
            raise FileNotFoundError(f"no avro files under {path}")
        pdf = avro_codec.read_avro(parts_all[0]).iloc[:0]
        return AnovosFrame.from_pandas(pdf, device=device)
    pdfs = [avro_codec.read_avro(p) for p in parts]
    pdf = pd.concat(pdfs, ignore_index=True) if len(pdfs) > 1 else pdfs[0]
    return AnovosFrame.from_pandas(pdf, device=device)


def write_dataset(idf: AnovosFrame, file_path: str, file_type: str, file_configs: Dict = None, column_order: List[str] = None):
    """Write the frame (all ranks write their shard as part files into a
    directory, Spark-style). mode: error|overwrite|append."""
    file_configs = dict(file_configs or {})
    mode = file_configs.get("mode", "error")
    ft = file_type.lower()
    if column_order:
        if len(column_order) != len(idf.columns):
            raise ValueError("Count of column(s) specified in column_order argument do not match Dataframe")
        diff = [x for x in column_order if x not in set(idf.columns)]
        if diff:
            raise ValueError(f"Column(s) specified in column_order argument not found in Dataframe: {diff}")
        idf = idf.select(column_order)
    if dist.rank() == 0:
        if os.path.exists(file_path):
            if mode == "error":
                raise FileExistsError(f"{file_path} exists (mode=error)")
            if mode == "overwrite":
                import shutil

                if os.path.isdir(file_path):
                    shutil.rmtree(file_path)
                else:
                    os.remove(file_path)
        os.makedirs(file_path, exist_ok=True)
    dist.barrier()
    part = os.path.join(file_path, f"part-{dist.rank():05d}")
    # vectorized materialization: categoricals stay dictionary-encoded
    # (parquet stores the dictionary; re-ingest is O(n)), timestamps stay
    # datetime64 — object-array churn dominated save/reread stages before
    pdf = idf.to_pandas_io() if ft in ("csv", "parquet", "json") else idf.to_pandas()
    if ft == "csv":
        header = str(file_configs.get("header", True)).lower() in ("true", "1")
        delim = file_configs.get("delimiter", ",")
        pdf.to_csv(part + ".csv", index=False, header=header, sep=delim)
    elif ft == "parquet":
        compression = file_configs.get("compression", "snappy")
        if compression == "uncompressed":
            compression = None
        papq.write_table(pa.Table.from_pandas(pdf, preserve_index=False), part + ".parquet", compression=compression)
    elif ft == "json":
        # ISO timestamps (Spark JSON parity; epoch-ms ints would come
        # back as plain numerics)
        pdf.to_json(part + ".json", orient="records", lines=True, date_format="iso")
    elif ft == "avro":
        from anovos_amd.core import avro_codec

        avro_codec.write_avro(pdf, part + ".avro")
    else:
        raise ValueError(f"unsupported file_type: {file_type}")
    dist.barrier()
