"""Minimal Avro Object Container File reader/writer (pure python).

The reference supports avro ingest via the spark-avro JAR
(data_ingest.py:36-38, shared/spark.py:85). No avro library ships in
this image, so this module implements the subset of the Avro 1.x spec
needed for dataset parity: records of primitive types and [null, T]
unions; codecs 'null' and 'deflate'.
"""

from __future__ import annotations

import io
import json
import os
import struct
import zlib
from typing import Any, Dict, List

import numpy as np
import pandas as pd

MAGIC = b"Obj\x01"


# ---------------- binary primitives ----------------
def _read_long(buf: io.BytesIO) -> int:
    shift = 0
    acc = 0
    while True:
        b = buf.read(1)
        if not b:
            raise EOFError
        byte = b[0]
        acc |= (byte & 0x7F) << shift
        if not (byte & 0x80):
            break
        shift += 7
    return (acc >> 1) ^ -(acc & 1)  # zigzag


def _write_long(out: io.BytesIO, n: int):
    n = (n << 1) ^ (n >> 63)
    while True:
        b = n & 0x7F
        n >>= 7
        if n:
            out.write(bytes([b | 0x80]))
        else:
            out.write(bytes([b]))
            break


def _read_bytes(buf: io.BytesIO) -> bytes:
    n = _read_long(buf)
    return buf.read(n)


def _write_bytes(out: io.BytesIO, b: bytes):
    _write_long(out, len(b))
    out.write(b)


def _read_value(buf: io.BytesIO, schema) -> Any:
    if isinstance(schema, list):  # union
        idx = _read_long(buf)
        return _read_value(buf, schema[idx])
    if isinstance(schema, dict):
        t = schema["type"]
        if isinstance(t, str) and t in ("int", "long", "float", "double", "boolean", "string", "bytes", "null") and "logicalType" in schema:
            return _read_value(buf, t)
        if t == "record":
            return {f["name"]: _read_value(buf, f["type"]) for f in schema["fields"]}
        if t in ("enum",):
            idx = _read_long(buf)
            return schema["symbols"][idx]
        if t == "array":
            out = []
            while True:
                n = _read_long(buf)
                if n == 0:
                    break
                if n < 0:
                    _read_long(buf)  # block size
                    n = -n
                for _ in range(n):
                    out.append(_read_value(buf, schema["items"]))
            return out
        return _read_value(buf, t)
    if schema == "null":
        return None
    if schema == "boolean":
        return buf.read(1)[0] != 0
    if schema in ("int", "long"):
        return _read_long(buf)
    if schema == "float":
        return struct.unpack("<f", buf.read(4))[0]
    if schema == "double":
        return struct.unpack("<d", buf.read(8))[0]
    if schema == "bytes":
        return _read_bytes(buf)
    if schema == "string":
        return _read_bytes(buf).decode("utf-8")
    raise NotImplementedError(f"avro type {schema}")


def _write_value(out: io.BytesIO, schema, v: Any):
    if isinstance(schema, list):
        if v is None:
            idx = schema.index("null")
            _write_long(out, idx)
            return
        idx = 0 if schema[0] != "null" else 1
        _write_long(out, idx)
        _write_value(out, schema[idx], v)
        return
    if isinstance(schema, dict) and schema["type"] == "record":
        for f in schema["fields"]:
            _write_value(out, f["type"], v[f["name"]])
        return
    if isinstance(schema, dict):  # logical type wrapper (timestamp-micros)
        _write_value(out, schema["type"], v)
        return
    if schema == "null":
        return
    if schema == "boolean":
        out.write(b"\x01" if v else b"\x00")
    elif schema in ("int", "long"):
        _write_long(out, int(v))
    elif schema == "float":
        out.write(struct.pack("<f", float(v)))
    elif schema == "double":
        out.write(struct.pack("<d", float(v)))
    elif schema == "string":
        _write_bytes(out, str(v).encode("utf-8"))
    elif schema == "bytes":
        _write_bytes(out, bytes(v))
    else:
        raise NotImplementedError(f"avro type {schema}")


# ---------------- container file ----------------
def read_avro(path: str) -> pd.DataFrame:
    with open(path, "rb") as f:
        data = f.read()
    buf = io.BytesIO(data)
    if buf.read(4) != MAGIC:
        raise ValueError(f"{path}: not an avro object container file")
    meta: Dict[str, bytes] = {}
    while True:
        n = _read_long(buf)
        if n == 0:
            break
        if n < 0:
            _read_long(buf)
            n = -n
        for _ in range(n):
            k = _read_bytes(buf).decode("utf-8")
            v = _read_bytes(buf)
            meta[k] = v
    schema = json.loads(meta["avro.schema"].decode("utf-8"))
    codec = meta.get("avro.codec", b"null").decode("utf-8")
    sync = buf.read(16)
    records: List[dict] = []
    while buf.tell() < len(data):
        try:
            cnt = _read_long(buf)
        except EOFError:
            break
        size = _read_long(buf)
        block = buf.read(size)
        if codec == "deflate":
            block = zlib.decompress(block, -15)
        elif codec != "null":
            raise NotImplementedError(f"avro codec {codec}")
        bb = io.BytesIO(block)
        for _ in range(cnt):
            records.append(_read_value(bb, schema))
        if buf.read(16) != sync:
            raise ValueError("avro sync marker mismatch")
    pdf = pd.DataFrame.from_records(records)
    # decode logical timestamp-micros columns
    for f in schema.get("fields", []):
        ftype = f["type"]
        parts = ftype if isinstance(ftype, list) else [ftype]
        for pt in parts:
            if isinstance(pt, dict) and pt.get("logicalType") == "timestamp-micros":
                name = f["name"]
                if name in pdf.columns:
                    pdf[name] = pd.to_datetime(pdf[name], unit="us")
    return pdf


def _schema_for(pdf: pd.DataFrame) -> dict:
    fields = []
    for name in pdf.columns:
        s = pdf[name]
        if pd.api.types.is_datetime64_any_dtype(s):
            t = {"type": "long", "logicalType": "timestamp-micros"}
        elif pd.api.types.is_float_dtype(s):
            t = "double"
        elif pd.api.types.is_integer_dtype(s):
            t = "long"
        elif pd.api.types.is_bool_dtype(s):
            t = "boolean"
        else:
            t = "string"
        fields.append({"name": str(name), "type": ["null", t]})
    return {"type": "record", "name": "anovos_row", "fields": fields}


def write_avro(pdf: pd.DataFrame, path: str, codec: str = "deflate"):
    schema = _schema_for(pdf)
    out = io.BytesIO()
    out.write(MAGIC)
    meta = {"avro.schema": json.dumps(schema).encode(), "avro.codec": codec.encode()}
    _write_long(out, len(meta))
    for k, v in meta.items():
        _write_bytes(out, k.encode())
        _write_bytes(out, v)
    _write_long(out, 0)
    sync = os.urandom(16)
    out.write(sync)
    body = io.BytesIO()
    cols = list(pdf.columns)
    n = len(pdf)
    arrs = {}
    for c in cols:
        if pd.api.types.is_datetime64_any_dtype(pdf[c]):
            arrs[c] = pdf[c].astype("datetime64[us]").astype("int64").to_numpy()
        else:
            arrs[c] = pdf[c].to_numpy()
    nulls = {c: pdf[c].isna().to_numpy() for c in cols}
    for i in range(n):
        rec = {c: (None if nulls[c][i] else arrs[c][i]) for c in cols}
        _write_value(body, schema, rec)
    payload = body.getvalue()
    if codec == "deflate":
        comp = zlib.compressobj(9, zlib.DEFLATED, -15)
        payload = comp.compress(payload) + comp.flush()
    _write_long(out, n)
    _write_long(out, len(payload))
    out.write(payload)
    out.write(sync)
    with open(path, "wb") as f:
        f.write(out.getvalue())
