#!/usr/bin/env python3
"""Feast repo codegen (reference notebook
feature_store__feast_exporter.ipynb): generate feature_definitions.py
from a frame + config and add the feast timestamp columns."""

import tempfile

from _common import AnovosFrame, demo_frame, init_context

from anovos_amd.feature_store import feast_exporter as fe

ctx = init_context()
idf = AnovosFrame.from_pandas(demo_frame(500), device=getattr(ctx, "device", "cpu"))
with tempfile.TemporaryDirectory() as td:
    cfg = {"entity": {"name": "customer", "id_col": "id", "description": "demo entity"},
           "file_source": {"owner": "demo@anovos", "description": "demo source",
                            "timestamp_col": "event_ts", "create_timestamp_col": "created_ts"},
           "feature_view": {"name": "customer_view", "ttl_in_seconds": 3600, "owner": "demo@anovos"},
           "file_path": td}
    fe.check_feast_configuration(cfg, 1)
    odf = fe.add_timestamp_columns(idf, cfg["file_source"])
    types = [(n, d) for n, d in odf.dtypes]
    fe.generate_feature_description(types, cfg, file_name="data.parquet")
    import os

    defs = [f for f in os.listdir(td) if f.endswith(".py")][0]
    print(open(os.path.join(td, defs)).read()[:600])
