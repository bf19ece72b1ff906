"""Shared helpers — parity with reference shared/utils.py.

- attributeType_segregation (reference shared/utils.py:48-73): split a
  frame's columns into (numerical, categorical, other) by dtype.
- get_dtype (:76), ends_with (:93), pairwise_reduce (:113),
  flatten_dataframe/transpose_dataframe analogs for tidy stats frames.
"""

from __future__ import annotations

from functools import reduce
from typing import List, Sequence, Tuple

import pandas as pd

from anovos_amd.core.dtypes import kind_of_dtype
from anovos_amd.core.frame import AnovosFrame


def attributeType_segregation(idf: AnovosFrame) -> Tuple[List[str], List[str], List[str]]:
    """Return (num_cols, cat_cols, other_cols) — the type system of the
    whole library (reference shared/utils.py:64-73)."""
    num_cols, cat_cols, other_cols = [], [], []
    for name, dtype in idf.dtypes:
        k = kind_of_dtype(dtype)
        if k == "numerical":
            num_cols.append(name)
        elif k == "categorical":
            cat_cols.append(name)
        else:
            other_cols.append(name)
    return num_cols, cat_cols, other_cols


def get_dtype(idf: AnovosFrame, col: str) -> str:
    """Reference shared/utils.py:76-90."""
    return dict(idf.dtypes)[col]


def ends_with(string: str, end_str: str = "/") -> str:
    """Ensure a path ends with end_str (reference shared/utils.py:93-110)."""
    string = str(string)
    if string.endswith(end_str):
        return string
    return string + end_str


def pairwise_reduce(op, x: Sequence):
    """Tree-reduction of a list (reference shared/utils.py:113-132) — kept
    for API parity; with tensor unions plain reduce is fine."""
    items = list(x)
    while len(items) > 1:
        nxt = []
        for i in range(0, len(items), 2):
            if i + 1 < len(items):
                nxt.append(op(items[i], items[i + 1]))
            else:
                nxt.append(items[i])
        items = nxt
    return items[0]


def normalize_columns(idf: AnovosFrame, list_of_cols, drop_cols=None, restrict_to=None) -> List[str]:
    """The cols/drop_cols normalization idiom used by every reference
    function (e.g. stats_generator.py:69-79): list or '|' string, 'all'
    sentinel, minus drop_cols; validates existence."""
    if list_of_cols in ("all", None):
        cols = restrict_to if restrict_to is not None else idf.columns
        cols = list(cols)
    elif isinstance(list_of_cols, str):
        cols = [c.strip() for c in list_of_cols.split("|") if c.strip()]
    else:
        cols = list(list_of_cols)
    if isinstance(drop_cols, str):
        drop_cols = [c.strip() for c in drop_cols.split("|") if c.strip()]
    drop_cols = set(drop_cols or [])
    cols = [c for c in cols if c not in drop_cols]
    seen = set()
    out = []
    for c in cols:
        if c not in seen:
            seen.add(c)
            out.append(c)
    missing = [c for c in out if c not in idf.columns]
    if missing:
        raise ValueError(f"columns not in frame: {missing}")
    return out


def union_stats(frames: List[pd.DataFrame]) -> pd.DataFrame:
    """Union tidy stats DataFrames (driver-side smalls)."""
    frames = [f for f in frames if f is not None and len(f)]
    if not frames:
        return pd.DataFrame()
    return pd.concat(frames, ignore_index=True)


def flatten_dataframe(idf, fixed_cols: List[str]) -> pd.DataFrame:
    """Melt/unpivot (reference shared/utils.py:6-25): every column NOT in
    fixed_cols becomes (key, value) long-format rows. Accepts a pandas
    frame or an AnovosFrame (stats tables are driver-side smalls)."""
    pdf = idf.to_pandas() if hasattr(idf, "to_pandas") else idf
    value_cols = [c for c in pdf.columns if c not in fixed_cols]
    return pdf.melt(id_vars=list(fixed_cols), value_vars=value_cols,
                    var_name="key", value_name="value")


def transpose_dataframe(idf, fixed_col: str) -> pd.DataFrame:
    """Transpose a stats frame about fixed_col (reference
    shared/utils.py:28-45: melt then pivot by fixed_col)."""
    pdf = idf.to_pandas() if hasattr(idf, "to_pandas") else idf
    out = pdf.set_index(fixed_col).T.reset_index().rename(columns={"index": fixed_col})
    out.columns.name = None
    return out


def output_to_local(output_path: str) -> str:
    """dbfs:/ -> /dbfs/ path munging (reference shared/utils.py:135-154);
    other schemes pass through."""
    p = str(output_path)
    if p.startswith("dbfs:"):
        return "/dbfs" + p[len("dbfs:"):]
    return p


def path_ak8s_modify(output_path: str, auth_key: str = "NA") -> str:
    """wasbs://container@account.blob.core.windows.net/p →
    https://account.blob.core.windows.net/container/p (reference
    shared/utils.py:157-179); non-wasbs paths pass through for local
    runs."""
    p = str(output_path)
    if not p.startswith("wasbs://"):
        return p
    container = p.split("//")[1].split("@")[0]
    rest = p.split("//")[1].split("@")[1]
    url = "https://" + rest.split("windows.net/")[0] + "windows.net"
    file_path_name = rest.split("windows.net/")[1]
    return url + "/" + container + "/" + file_path_name


def cloud_sync(local_path: str, master_path: str, run_type: str = "local", auth_key: str = "NA", recursive: bool = False):
    """Reference parity for the emr/ak8s side-channel copies
    (report_preprocessing.py:97-119, ts_analyzer.py:452-460): shells
    ``aws s3 cp`` / ``azcopy`` to push locally-written stats to cloud
    storage. No-op for local/databricks (databricks paths are direct
    via output_to_local)."""
    import subprocess

    if run_type == "emr":
        tgt = ends_with(master_path) if recursive else master_path
        cmd = ["aws", "s3", "cp"] + (["--recursive"] if recursive else []) + [ends_with(local_path) if recursive else local_path, tgt]
        subprocess.check_output(cmd)
    elif run_type == "ak8s":
        base = ends_with(path_ak8s_modify(master_path)) if recursive else path_ak8s_modify(master_path)
        target = base + (str(auth_key) if auth_key != "NA" else "")
        cmd = ["azcopy", "cp", local_path, target] + (["--recursive=true"] if recursive else [])
        subprocess.check_output(cmd)
