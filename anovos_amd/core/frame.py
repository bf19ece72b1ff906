"""AnovosFrame — the engine's DataFrame: a column store of torch tensors.

Replaces the reference's Spark DataFrame (every reference function takes
``(spark, idf, ...)`` and returns DataFrames). Here a frame is a dict of
named ``Column`` objects, each one a contiguous torch tensor living on the
local GPU (HBM3E-resident) or CPU, holding this rank's row shard.
Cross-rank semantics: frames are row-partitioned; global statistics are
produced by the ops layer via RCCL partial-aggregate merges
(core/dist.py), never by shuffling rows.

Columns (core/dtypes.py):
  numerical   float32/float64 tensor, NaN null
  categorical int32 codes + host dictionary (list[str]), -1 null
  timestamp   int64 epoch-microseconds, INT64_MIN null
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Sequence, Union

import numpy as np
import pandas as pd
import torch

from anovos_amd.core.dtypes import NULL_CODE, NULL_TS, is_timestamp_dtype, kind_of_dtype


class Column:
    """One column: tensor data + logical dtype + optional dictionary.

    ``cache`` holds derived statistics (moments vector, histograms, null
    counts) keyed by the ops layer — the in-memory analog of the
    reference's pre-saved stats reuse (workflow.stats_args, reference
    workflow.py:91-145). A new Column (new tensor) starts cold; sharing a
    Column between frames shares its cache."""

    __slots__ = ("name", "dtype", "data", "dictionary", "cache")

    def __init__(self, name: str, dtype: str, data: torch.Tensor, dictionary: Optional[List[str]] = None):
        self.name = name
        self.dtype = dtype  # Spark-style dtype string ('double', 'string', 'int', ...)
        self.data = data
        self.dictionary = dictionary
        self.cache = {}

    @property
    def kind(self) -> str:
        return kind_of_dtype(self.dtype)

    def null_mask(self) -> torch.Tensor:
        """Boolean tensor, True where null."""
        if self.kind == "numerical":
            return torch.isnan(self.data)
        if self.kind == "categorical":
            return self.data == NULL_CODE
        if is_timestamp_dtype(self.dtype):
            return self.data == NULL_TS
        return torch.zeros_like(self.data, dtype=torch.bool)

    def clone(self) -> "Column":
        return Column(self.name, self.dtype, self.data, list(self.dictionary) if self.dictionary else None)

    def gather(self, idx: torch.Tensor) -> "Column":
        return Column(self.name, self.dtype, self.data[idx], self.dictionary)

    def to_numpy_objects(self) -> np.ndarray:
        """Materialize as a numpy array with python objects/NaN for report/IO paths."""
        t = self.data.cpu()
        if self.kind == "categorical":
            codes = t.numpy()
            d = np.array(self.dictionary + [None], dtype=object) if self.dictionary else np.array([None], dtype=object)
            out = d[np.where(codes == NULL_CODE, len(d) - 1, codes)]
            return out
        if is_timestamp_dtype(self.dtype):
            v = t.numpy()
            out = v.astype("datetime64[us]").astype(object)
            out = np.where(v == NULL_TS, None, out)
            return out
        return t.numpy()

    def to_pandas_series(self) -> pd.Series:
        """Vectorized pandas materialization for IO paths: categorical
        columns come back as pd.Categorical (parquet then stores the
        dictionary, and re-ingest takes the O(n) Categorical fast path
        instead of re-factorizing object arrays), timestamps as
        datetime64[us] with NaT."""
        t = self.data.cpu()
        if self.kind == "categorical":
            codes = t.numpy().astype(np.int64, copy=True)
            codes[codes == NULL_CODE] = -1
            cats = pd.Index([str(x) for x in (self.dictionary or [])], dtype=object)
            return pd.Series(pd.Categorical.from_codes(codes, categories=cats))
        if is_timestamp_dtype(self.dtype):
            v = t.numpy()
            out = v.astype("datetime64[us]")
            out[v == NULL_TS] = np.datetime64("NaT")
            return pd.Series(out)
        return pd.Series(t.numpy())


class AnovosFrame:
    """Row-sharded column store. All mutating ops return new frames
    (tensors are shared where unchanged — cheap copy-on-write)."""

    def __init__(self, columns: Optional[Dict[str, Column]] = None, device: Union[str, torch.device] = "cpu"):
        self._cols: Dict[str, Column] = dict(columns or {})
        self.device = torch.device(device)

    # ---------------- basic introspection ----------------
    @property
    def columns(self) -> List[str]:
        return list(self._cols.keys())

    @property
    def dtypes(self) -> List[tuple]:
        return [(c.name, c.dtype) for c in self._cols.values()]

    def col(self, name: str) -> Column:
        if name not in self._cols:
            raise KeyError(f"column '{name}' not in frame (have {self.columns})")
        return self._cols[name]

    def __contains__(self, name: str) -> bool:
        return name in self._cols

    def __len__(self) -> int:
        return self.local_rows()

    def local_rows(self) -> int:
        for c in self._cols.values():
            return int(c.data.shape[0])
        return 0

    def count(self) -> int:
        """Global row count across ranks (RCCL all-reduce when
        distributed). Cached — a frame's row count is immutable (every
        row-changing op returns a new frame)."""
        if getattr(self, "_count_cache", None) is None:
            from anovos_amd.core import dist

            self._count_cache = dist.all_reduce_scalar(self.local_rows())
        return self._count_cache

    # ---------------- construction ----------------
    @staticmethod
    def from_pandas(pdf: pd.DataFrame, device="cpu", dtype_overrides: Optional[Dict[str, str]] = None) -> "AnovosFrame":
        cols: Dict[str, Column] = {}
        dev = torch.device(device)
        for name in pdf.columns:
            s = pdf[name]
            cols[str(name)] = _column_from_series(str(name), s, dev, (dtype_overrides or {}).get(str(name)))
        return AnovosFrame(cols, dev)

    def to_pandas(self) -> pd.DataFrame:
        """Local shard to pandas (driver-side smalls only — stats frames etc.)."""
        data = {}
        for name, c in self._cols.items():
            data[name] = c.to_numpy_objects()
        return pd.DataFrame(data)

    def to_pandas_io(self) -> pd.DataFrame:
        """Vectorized pandas materialization for dataset writes
        (categoricals stay dictionary-encoded, timestamps stay
        datetime64) — O(n) with no python-object churn."""
        return pd.DataFrame({name: c.to_pandas_series() for name, c in self._cols.items()})

    # ---------------- column ops (reference data_ingest.py:201-367 semantics) ----------------
    def select(self, names: Sequence[str]) -> "AnovosFrame":
        return AnovosFrame({n: self._cols[n] for n in names}, self.device)

    def drop(self, names: Iterable[str]) -> "AnovosFrame":
        names = set(names)
        return AnovosFrame({n: c for n, c in self._cols.items() if n not in names}, self.device)

    def rename(self, mapping: Dict[str, str]) -> "AnovosFrame":
        out = {}
        for n, c in self._cols.items():
            nn = mapping.get(n, n)
            cc = c.clone()
            cc.name = nn
            out[nn] = cc
        return AnovosFrame(out, self.device)

    def with_column(self, name: str, col: Column) -> "AnovosFrame":
        out = dict(self._cols)
        col.name = name
        out[name] = col
        return AnovosFrame(out, self.device)

    def filter_rows(self, keep: torch.Tensor) -> "AnovosFrame":
        """keep: boolean mask or index tensor over local rows."""
        if keep.dtype == torch.bool:
            idx = keep.nonzero(as_tuple=True)[0]
        else:
            idx = keep
        return AnovosFrame({n: c.gather(idx) for n, c in self._cols.items()}, self.device)

    def cast(self, name: str, new_dtype: str) -> "AnovosFrame":
        c = self._cols[name]
        nd = new_dtype.lower()
        new_kind = kind_of_dtype(nd)
        if new_kind == "numerical":
            tdt = torch.float64 if nd in ("double", "bigint", "long") else torch.float32
            if c.kind == "numerical":
                data = c.data.to(tdt)
                newc = Column(name, nd, data)
            elif c.kind == "categorical":
                # parse dictionary entries as floats; unparseable -> NaN
                vals = []
                for s in c.dictionary or []:
                    try:
                        vals.append(float(s))
                    except (TypeError, ValueError):
                        vals.append(float("nan"))
                lut = torch.tensor(vals + [float("nan")], dtype=tdt, device=c.data.device)
                codes = c.data.to(torch.long)
                codes = torch.where(codes == NULL_CODE, torch.full_like(codes, len(vals)), codes)
                newc = Column(name, nd, lut[codes])
            else:
                newc = Column(name, nd, c.data.to(tdt))
        elif new_kind == "categorical":
            if c.kind == "categorical":
                newc = Column(name, "string", c.data, c.dictionary)
            else:
                from anovos_amd.ops import encode

                newc = encode.numeric_to_string_column(c)
        else:
            newc = Column(name, nd, c.data, c.dictionary)
        return self.with_column(name, newc)

    def to_device(self, device) -> "AnovosFrame":
        dev = torch.device(device)
        out = {n: Column(c.name, c.dtype, c.data.to(dev), c.dictionary) for n, c in self._cols.items()}
        return AnovosFrame(out, dev)

    @property
    def aux_cache(self) -> dict:
        """Frame-instance-scoped cache for derived aggregates that span
        multiple columns (e.g. IV/IG shared binned label counts). Not
        inherited by derived frames (with_column/select return new
        instances), cleared with clear_stats_cache."""
        d = getattr(self, "_aux_cache", None)
        if d is None:
            d = {}
            self._aux_cache = d
        return d

    def clear_stats_cache(self) -> None:
        """Drop all cached derived statistics (forces recomputation —
        used by benchmarks to keep timed steps honest)."""
        for c in self._cols.values():
            c.cache.clear()
        self._aux_cache = {}

    def persist(self) -> "AnovosFrame":  # Spark-parity no-op: tensors are already resident
        return self

    def unpersist(self) -> "AnovosFrame":
        return self

    def copy(self) -> "AnovosFrame":
        return AnovosFrame({n: c.clone() for n, c in self._cols.items()}, self.device)

    def __repr__(self):
        return f"AnovosFrame({self.local_rows()} local rows x {len(self._cols)} cols on {self.device})"


def _column_from_series(name: str, s: pd.Series, dev: torch.device, override: Optional[str] = None) -> Column:
    """Build a Column from a pandas Series, inferring the Spark-style dtype."""
    if override:
        k = kind_of_dtype(override)
        if k == "categorical":
            return _string_column(name, s, dev)
        if k == "numerical":
            arr = pd.to_numeric(s, errors="coerce").astype("float64" if override in ("double", "bigint", "long") else "float32")
            t = torch.from_numpy(np.ascontiguousarray(arr.to_numpy())).to(dev)
            return Column(name, override, t)
    if pd.api.types.is_datetime64_any_dtype(s):
        vals = s.astype("datetime64[us]")
        arr = vals.to_numpy().astype("int64")
        arr = np.where(s.isna().to_numpy(), NULL_TS, arr)
        return Column(name, "timestamp", torch.from_numpy(arr).to(dev))
    if pd.api.types.is_float_dtype(s):
        arr = s.to_numpy(dtype="float64" if s.dtype == np.float64 else "float32")
        dt = "double" if arr.dtype == np.float64 else "float"
        return Column(name, dt, torch.from_numpy(np.ascontiguousarray(arr)).to(dev))
    if pd.api.types.is_bool_dtype(s):
        arr = s.to_numpy(dtype="float32")
        return Column(name, "int", torch.from_numpy(arr).to(dev))
    if pd.api.types.is_integer_dtype(s):
        if s.isna().any():
            arr = s.astype("float64").to_numpy()
        else:
            arr = s.to_numpy()
        wide = arr.dtype.itemsize > 4 if arr.dtype.kind in "iu" else True
        out = arr.astype("float64" if wide else "float32")
        dt = "bigint" if wide else "int"
        return Column(name, dt, torch.from_numpy(np.ascontiguousarray(out)).to(dev))
    return _string_column(name, s, dev)


def _string_column(name: str, s: pd.Series, dev: torch.device) -> Column:
    # fast path: pandas Categorical (parquet dictionary round-trip) —
    # remap codes to the SORTED dictionary the engine standardizes on
    if isinstance(s.dtype, pd.CategoricalDtype):
        cats = np.asarray(s.cat.categories.astype(str))
        order = np.argsort(cats)
        remap = np.empty(len(cats), dtype=np.int32)
        remap[order] = np.arange(len(cats), dtype=np.int32)
        codes = s.cat.codes.to_numpy()
        out = np.where(codes >= 0, remap[np.maximum(codes, 0)], NULL_CODE).astype(np.int32)
        return Column(name, "string", torch.from_numpy(out).to(dev), [str(c) for c in cats[order]])
    mask = s.isna().to_numpy()
    filled = s.where(~s.isna(), "").astype(str)
    # pd.factorize(sort=True): C-speed hashing, codes against the sorted
    # dictionary (same order np.unique produced; cross-rank unify at
    # ingest handles the rest). The old per-element python loop was the
    # dominant cost of every save/reread materialization (~9 s per stage
    # at 1M rows x 10 string cols).
    codes, uniq = pd.factorize(filled, sort=True)
    codes = codes.astype(np.int32)
    codes[mask] = NULL_CODE
    return Column(name, "string", torch.from_numpy(codes).to(dev), [str(u) for u in uniq])


def _dict_encode(arr: np.ndarray):
    """Dictionary-encode an object array -> (int32 codes, list[str]).
    Sorted dictionary order; null -> NULL_CODE."""
    s = pd.Series(arr, dtype=object)
    mask = s.isna().to_numpy()
    filled = s.where(~s.isna(), "").astype(str)
    codes, uniq = pd.factorize(filled, sort=True)
    codes = codes.astype(np.int32)
    codes[mask] = NULL_CODE
    return codes, [str(u) for u in uniq]
