"""Time-series auto-detection + analyzer tests (reference parity:
src/test/anovos ts tests; inline frames, hand-computed expectations)."""

import os

import numpy as np
import pandas as pd
import pytest
import torch

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_ingest import ts_auto_detection as tsd
from anovos_amd.data_analyzer import ts_analyzer as tsa
from anovos_amd.shared.context import init_context


@pytest.fixture
def ctx():
    return init_context("cpu")


def test_regex_parser_string_dates(ctx):
    pdf = pd.DataFrame({"d": ["2020-01-01", "2020-06-15", "2021-03-03", None]})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = tsd.regex_date_time_parser(ctx, idf, "d")
    c = odf.col("d")
    assert c.dtype == "timestamp"
    assert list(c.null_mask().numpy()) == [False, False, False, True]


def test_regex_parser_epoch10(ctx):
    pdf = pd.DataFrame({"e": [1577836800, 1592179200, 1614729600]})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = tsd.regex_date_time_parser(ctx, idf, "e")
    c = odf.col("e")
    assert c.dtype == "timestamp"
    assert int(c.data[0]) == 1577836800 * 1_000_000


def test_regex_parser_yyyymmdd_int(ctx):
    pdf = pd.DataFrame({"ymd": [20200101, 20200615, 20210303]})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = tsd.regex_date_time_parser(ctx, idf, "ymd")
    c = odf.col("ymd")
    assert c.dtype == "timestamp"
    d0 = pd.Timestamp(int(c.data[0]), unit="us")
    assert (d0.year, d0.month, d0.day) == (2020, 1, 1)


def test_regex_parser_leaves_words(ctx):
    pdf = pd.DataFrame({"w": ["apple", "banana", "cherry"]})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = tsd.regex_date_time_parser(ctx, idf, "w")
    assert odf.col("w").dtype == "string"


def test_ts_loop_cols_pre(ctx):
    pdf = pd.DataFrame(
        {
            "id": range(10),
            "date_str": ["2020-01-%02d" % (i + 1) for i in range(10)],
            "epoch": [1577836800 + i * 86400 for i in range(10)],
            "word": ["foobarbaz"] * 10,  # fixed length 9 — not a candidate width
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    lc1, lc2, lc3 = tsd.ts_loop_cols_pre(idf, "id")
    m = dict(zip(lc1, lc2))
    assert m["date_str"] == "string_c"  # fixed length 10
    assert m["epoch"] in ("int_c", "bigint_c")
    assert m["word"] == "string"


def test_ts_preprocess(ctx, tmp_path):
    pdf = pd.DataFrame(
        {
            "id": range(20),
            "d": ["2020-%02d-01" % ((i % 12) + 1) for i in range(20)],
            "x": np.random.default_rng(0).normal(size=20),
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf, ts_cols, num_cols, cat_cols = tsd.ts_preprocess(ctx, idf, "id", str(tmp_path))
    assert "d" in ts_cols
    assert "x" in num_cols
    stats = pd.read_csv(tmp_path / "ts_cols_stats.csv")
    assert "d" in list(stats["attribute"])


def test_daypart_cat_scalar():
    assert tsa.daypart_cat(5) == "early_hours"
    assert tsa.daypart_cat(12) == "work_hours"
    assert tsa.daypart_cat(23) == "late_hours"
    assert tsa.daypart_cat(8) == "commuting_hours"
    assert tsa.daypart_cat(21) == "other_hours"
    assert tsa.daypart_cat(None) == "Missing_NA"


def test_ts_analyzer_outputs(ctx, tmp_path):
    rng = np.random.default_rng(1)
    n = 200
    pdf = pd.DataFrame(
        {
            "id": rng.integers(0, 20, n),
            "ts": pd.to_datetime("2020-01-01") + pd.to_timedelta(rng.integers(0, 90, n), unit="D")
            + pd.to_timedelta(rng.integers(0, 24, n), unit="h"),
            "amount": rng.normal(100, 10, n),
            "cat": pd.Series(rng.choice(["a", "b", "c"], n)),
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    ts_cols = tsa.ts_analyzer(ctx, idf, "id", 90, str(tmp_path))
    assert ts_cols == ["ts"]
    assert os.path.exists(tmp_path / "stats_ts_1.csv")
    assert os.path.exists(tmp_path / "stats_ts_2.csv")
    f1 = pd.read_csv(tmp_path / "stats_ts_1.csv")
    assert f1["mean"][0] == pytest.approx(1.0, abs=0.3)  # near-daily coverage
    viz = pd.read_csv(tmp_path / "ts_amount_daily.csv")
    assert {"min", "max", "mean", "median"} <= set(viz.columns)
    catviz = pd.read_csv(tmp_path / "ts_cat_daily.csv")
    assert set(catviz["cat"]) <= {"a", "b", "c"}


def test_ts_viz_weekly_and_hourly(ctx):
    rng = np.random.default_rng(2)
    n = 500
    pdf = pd.DataFrame(
        {
            "id": rng.integers(0, 50, n),
            "ts": pd.to_datetime("2020-01-01") + pd.to_timedelta(rng.integers(0, 60, n), unit="D")
            + pd.to_timedelta(rng.integers(0, 24, n), unit="h"),
            "v": rng.normal(size=n),
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    feats = tsa.ts_processed_feats(idf, "ts", "id")
    wk = tsa.ts_viz_data(feats, "ts", "v", output_type="weekly")
    assert len(wk) <= 7
    hr = tsa.ts_viz_data(feats, "ts", "v", output_type="hourly")
    assert set(hr["daypart_cat"]) <= set(tsa.DAYPARTS)
