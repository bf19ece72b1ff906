"""Feast historical-retrieval demo (reference parity:
anovos/feature_store/feature_retrieval.py:8-58).

Reads back point-in-time-correct features from the Feast repo that
``feast_exporter.generate_feature_description`` emitted (the
``write_feast_features`` workflow stage). ``feast`` is an optional
dependency — absent in this offline image — so the demo degrades to a
direct parquet read of the exported file source when the package is
missing, keeping the join semantics (latest feature row at or before
each entity's event_time) so the demo remains runnable end-to-end.
"""

from __future__ import annotations

import os
import sys
from datetime import datetime

import pandas as pd

try:
    import feast  # type: ignore

    HAS_FEAST = True
except ImportError:  # pragma: no cover - optional dependency
    feast = None
    HAS_FEAST = False


def _demo_entities(ids):
    now = datetime.now()
    return pd.DataFrame({"ifa": list(ids), "event_time": [now] * len(ids)})


def _fallback_historical(repo_path: str, entity_df: pd.DataFrame, feature_cols):
    """Offline stand-in for fs.get_historical_features: point-in-time
    join against the exported parquet source (latest row whose
    timestamp <= event_time per entity)."""
    import glob

    src = sorted(glob.glob(os.path.join(repo_path, "**", "*.parquet"), recursive=True))
    if not src:
        raise FileNotFoundError(f"no parquet file source under {repo_path}")
    feat = pd.concat([pd.read_parquet(p) for p in src], ignore_index=True)
    ts_col = "event_time" if "event_time" in feat.columns else None
    out_rows = []
    for _, e in entity_df.iterrows():
        rows = feat[feat["ifa"] == e["ifa"]]
        if ts_col:
            rows = rows[pd.to_datetime(rows[ts_col]) <= pd.to_datetime(e["event_time"])]
            rows = rows.sort_values(ts_col)
        if len(rows):
            r = rows.iloc[-1]
            out_rows.append([e["ifa"], e["event_time"]] + [r.get(c) for c in feature_cols])
        else:
            out_rows.append([e["ifa"], e["event_time"]] + [None] * len(feature_cols))
    return pd.DataFrame(out_rows, columns=["ifa", "event_time"] + list(feature_cols))


def retrieve_historical_feature_demo(repo_path: str, entity_ids=None, features=None,
                                     service_name: str = "income_feature_service"):
    """Reference feature_retrieval.py:8 — historical retrieval by
    explicit feature refs and by feature service."""
    ids = entity_ids or ["27a", "30a", "475a", "965a", "1678a", "1698a", "1807a", "1951a", "2041a", "2215a"]
    entities = _demo_entities(ids)
    features = features or ["income_view:income"]

    if HAS_FEAST:
        fs = feast.FeatureStore(repo_path=repo_path)
        df = fs.get_historical_features(entity_df=entities, features=features).to_df()
        print(df.head())
        feature_service = fs.get_feature_service(service_name)
        df2 = fs.get_historical_features(features=feature_service, entity_df=entities).to_df()
        print(df2.head())
        return df, df2

    cols = [f.split(":", 1)[1] for f in features]
    df = _fallback_historical(repo_path, entities, cols)
    print(df.head())
    return df, df


if __name__ == "__main__":
    if len(sys.argv) < 2:
        print("Please, provide a path to anovos feature repo!")
        sys.exit(1)
    retrieve_historical_feature_demo(repo_path=sys.argv[1])
