#!/usr/bin/env python3
"""Generate a synthetic income-style dataset (csv + parquet) for the
example configs under config/ (the reference shipped a real income
dataset under data/; no datasets are fetchable in this stack)."""

import argparse
import os

import numpy as np
import pandas as pd


def make(n: int, seed: int = 11) -> pd.DataFrame:
    rng = np.random.default_rng(seed)
    edu = rng.choice(["HS-grad", "Some-college", "Bachelors", "Masters", "Doctorate"], n,
                     p=[0.35, 0.25, 0.22, 0.13, 0.05])
    edu_num = pd.Series(edu).map({"HS-grad": 9, "Some-college": 10, "Bachelors": 13,
                                  "Masters": 14, "Doctorate": 16}).to_numpy()
    age = rng.integers(17, 90, n)
    hours = np.clip(rng.normal(40, 12, n), 1, 99).round()
    gain = np.where(rng.random(n) < 0.92, 0, rng.integers(114, 99999, n))
    loss = np.where(rng.random(n) < 0.95, 0, rng.integers(155, 4356, n))
    score = (age / 90 + edu_num / 16 + hours / 99 + (gain > 0) * 0.8 + rng.normal(0, 0.35, n))
    income = np.where(score > 1.75, ">50K", "<=50K")
    df = pd.DataFrame(
        {
            "ifa": [f"27520a:{i:07d}" for i in range(n)],
            "age": age.astype(float),
            "workclass": rng.choice(["Private", "Self-emp-not-inc", "Local-gov", "State-gov", "Federal-gov"], n,
                                    p=[0.7, 0.1, 0.08, 0.07, 0.05]),
            "fnlwgt": rng.integers(12285, 1484705, n),
            "logfnl": rng.normal(11.7, 0.6, n),
            "education": edu,
            "education-num": edu_num.astype(float),
            "marital-status": rng.choice(["Married-civ-spouse", "Never-married", "Divorced", "Widowed"], n,
                                         p=[0.46, 0.33, 0.15, 0.06]),
            "occupation": rng.choice(["Prof-specialty", "Craft-repair", "Exec-managerial", "Adm-clerical",
                                      "Sales", "Other-service"], n),
            "relationship": rng.choice(["Husband", "Not-in-family", "Own-child", "Unmarried", "Wife"], n,
                                       p=[0.4, 0.26, 0.15, 0.11, 0.08]),
            "race": rng.choice(["White", "Black", "Asian-Pac-Islander", "Other"], n, p=[0.85, 0.1, 0.03, 0.02]),
            "sex": rng.choice(["Male", "Female"], n, p=[0.67, 0.33]),
            "capital-gain": gain.astype(float),
            "capital-loss": loss.astype(float),
            "hours-per-week": hours,
            "native-country": rng.choice(["United-States", "Mexico", "Philippines", "Germany", "India"], n,
                                         p=[0.9, 0.04, 0.02, 0.02, 0.02]),
            "income": income,
        }
    )
    # inject some nulls
    for c in ("workclass", "occupation", "native-country"):
        df.loc[rng.random(n) < 0.05, c] = None
    df.loc[rng.random(n) < 0.01, "age"] = np.nan
    return df


def add_geo_cols(df: pd.DataFrame, seed: int = 21) -> pd.DataFrame:
    """Append latitude/longitude (US-ish) and a geohash column."""
    import sys, os
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from anovos_amd.data_ingest.geo_auto_detection import latlong_to_geo

    rng = np.random.default_rng(seed)
    n = len(df)
    lat = rng.uniform(25.5, 48.5, n)
    lon = rng.uniform(-123.0, -68.0, n)
    df = df.copy()
    df["latitude"] = lat
    df["longitude"] = lon
    df["gh7"] = latlong_to_geo(lat, lon, precision=7)
    return df


def add_ts_cols(df: pd.DataFrame, seed: int = 31) -> pd.DataFrame:
    """Append transaction-date / signup-epoch timestamp candidates."""
    rng = np.random.default_rng(seed)
    n = len(df)
    base = pd.Timestamp("2020-01-01")
    df = df.copy()
    df["txn_date"] = (base + pd.to_timedelta(rng.integers(0, 365, n), unit="D")
                      + pd.to_timedelta(rng.integers(0, 86400, n), unit="s")).strftime("%Y-%m-%d %H:%M:%S")
    df["signup_epoch"] = (1577836800 + rng.integers(0, 365 * 86400, n)).astype("int64")
    return df


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--rows", type=int, default=32561)
    ap.add_argument("--out", default="data/income_dataset")
    ap.add_argument("--geo", action="store_true", help="append lat/lon/geohash columns")
    ap.add_argument("--ts", action="store_true", help="append timestamp candidate columns")
    ap.add_argument("--snapshots", type=int, default=0, help="write N drifting snapshot datasets")
    a = ap.parse_args()
    df = make(a.rows)
    if a.geo:
        df = add_geo_cols(df)
    if a.ts:
        df = add_ts_cols(df)
    os.makedirs(os.path.join(a.out, "csv"), exist_ok=True)
    os.makedirs(os.path.join(a.out, "parquet"), exist_ok=True)
    df.to_csv(os.path.join(a.out, "csv", "part-00000.csv"), index=False)
    df.to_parquet(os.path.join(a.out, "parquet", "part-00000.parquet"))
    # data dictionary for the report Wiki tab (reference ships one per
    # example dataset: examples/data/income_dataset/data_dictionary.csv)
    definitions = {
        "ifa": "unique row identifier.",
        "age": "continuous.",
        "workclass": "employment class (Private, Self-emp, Gov, ...).",
        "fnlwgt": "continuous sampling weight.",
        "logfnl": "log of fnlwgt.",
        "education": "highest education level attained.",
        "education-num": "continuous education rank.",
        "marital-status": "marital status category.",
        "occupation": "occupation category.",
        "relationship": "household relationship category.",
        "race": "race category.",
        "sex": "sex category.",
        "capital-gain": "continuous.",
        "capital-loss": "continuous.",
        "hours-per-week": "continuous working hours.",
        "native-country": "country of origin.",
        "income": "label: <=50K or >50K.",
        "latitude": "location latitude (decimal degrees).",
        "longitude": "location longitude (decimal degrees).",
        "gh7": "precision-7 geohash of (latitude, longitude).",
        "txn_date": "transaction timestamp string.",
        "signup_epoch": "signup time, 10-digit epoch seconds.",
    }
    pd.DataFrame({"attribute": [c for c in df.columns],
                  "definition": [definitions.get(c, "") for c in df.columns]}
                 ).to_csv(os.path.join(a.out, "data_dictionary.csv"), index=False)
    # drift source: mildly shifted snapshot
    src = make(a.rows, seed=12)
    src["age"] = src["age"] * 1.05
    os.makedirs(os.path.join(a.out, "source", "csv"), exist_ok=True)
    src.to_csv(os.path.join(a.out, "source", "csv", "part-00000.csv"), index=False)
    # join dataset (avro): ifa + duplicated demo columns (reference
    # config joins an avro side table, configs.yaml:51-63)
    import sys as _sys
    _sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    from anovos_amd.core.avro_codec import write_avro

    join_df = df[["ifa", "age", "workclass"]].rename(columns={"age": "dupl_age", "workclass": "dupl_workclass"})
    os.makedirs(os.path.join(a.out, "join"), exist_ok=True)
    write_avro(join_df, os.path.join(a.out, "join", "part-00000.avro"))
    for k in range(a.snapshots):
        snap = make(max(a.rows // 4, 1000), seed=100 + k)
        # gradual drift across snapshots
        snap["age"] = snap["age"] * (1 + 0.01 * k)
        snap["hours-per-week"] = snap["hours-per-week"] + 0.3 * k
        d = os.path.join(a.out, f"snapshot{k + 1:02d}", "csv")
        os.makedirs(d, exist_ok=True)
        snap.to_csv(os.path.join(d, "part-00000.csv"), index=False)
    print("wrote", a.out)
