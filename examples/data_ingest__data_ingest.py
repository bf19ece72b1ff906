#!/usr/bin/env python3
"""Ingestion + basic ETL (reference notebook data_ingest__data_ingest.ipynb):
read/write round-trips across formats, concat, join, recast,
recommend_type."""

import tempfile

from _common import AnovosFrame, demo_frame, init_context

from anovos_amd.data_ingest import data_ingest as di

ctx = init_context()
pdf = demo_frame(2000)
idf = AnovosFrame.from_pandas(pdf, device=getattr(ctx, "device", "cpu"))
with tempfile.TemporaryDirectory() as td:
    for ft in ("csv", "parquet", "json", "avro"):
        di.write_dataset(idf, f"{td}/ds_{ft}", ft, {"header": True, "mode": "overwrite"})
        back = di.read_dataset(ctx, f"{td}/ds_{ft}", ft, {"header": True, "inferSchema": True})
        print(ft, "roundtrip rows:", back.count())
half = AnovosFrame.from_pandas(pdf.iloc[:1000], device=getattr(ctx, "device", "cpu"))
print("concat:", di.concatenate_dataset(half, half, method_type="name").count())
left = di.select_column(idf, ["id", "age"])
right = di.select_column(idf, ["id", "income"])
print("join:", di.join_dataset(left, right, join_cols="id", join_type="inner").count())
print(di.recommend_type(ctx, idf, drop_cols=["id"]).to_string(index=False))
