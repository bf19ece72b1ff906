import numpy as np
import pandas as pd
import pytest

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_ingest import data_ingest as di


@pytest.fixture()
def small_pdf():
    return pd.DataFrame(
        {
            "ifa": ["a", "b", "c", "d"],
            "x": [1.0, 2.0, np.nan, 4.0],
            "y": ["u", "v", "u", None],
        }
    )


@pytest.mark.parametrize("ft", ["csv", "parquet", "json", "avro"])
def test_write_read_roundtrip(ctx, small_pdf, tmp_path, ft):
    f = AnovosFrame.from_pandas(small_pdf)
    path = str(tmp_path / f"ds_{ft}")
    di.write_dataset(f, path, ft, {"mode": "overwrite", "header": True})
    back = di.read_dataset(ctx, path, ft)
    pdf = back.to_pandas().sort_values("ifa").reset_index(drop=True)
    assert list(pdf["ifa"]) == ["a", "b", "c", "d"]
    assert pdf["x"][3] == 4.0
    assert pd.isna(pdf["x"][2])
    assert pdf["y"][0] == "u"
    assert pdf["y"][3] is None or pd.isna(pdf["y"][3])


def test_concatenate_by_name(small_pdf):
    a = AnovosFrame.from_pandas(small_pdf)
    b = AnovosFrame.from_pandas(small_pdf[["y", "x", "ifa"]])  # shuffled cols
    out = di.concatenate_dataset(a, b, method_type="name")
    assert out.local_rows() == 8
    assert out.columns == a.columns
    pdf = out.to_pandas()
    assert list(pdf["ifa"][:4]) == list(pdf["ifa"][4:])


def test_concatenate_by_index(small_pdf):
    a = AnovosFrame.from_pandas(small_pdf)
    b = AnovosFrame.from_pandas(small_pdf)
    out = di.concatenate_dataset(a, b, method_type="index")
    assert out.local_rows() == 8


def test_join_inner_left(small_pdf):
    a = AnovosFrame.from_pandas(small_pdf)
    right = pd.DataFrame({"ifa": ["a", "c", "z"], "z": [10.0, 30.0, 99.0]})
    b = AnovosFrame.from_pandas(right)
    out = di.join_dataset(a, b, join_cols="ifa", join_type="inner")
    pdf = out.to_pandas().sort_values("ifa").reset_index(drop=True)
    assert list(pdf["ifa"]) == ["a", "c"]
    assert list(pdf["z"]) == [10.0, 30.0]
    out2 = di.join_dataset(a, b, join_cols="ifa", join_type="left")
    pdf2 = out2.to_pandas().sort_values("ifa").reset_index(drop=True)
    assert len(pdf2) == 4
    assert pd.isna(pdf2["z"][1])  # 'b' unmatched
    anti = di.join_dataset(a, b, join_cols="ifa", join_type="left_anti")
    assert sorted(anti.to_pandas()["ifa"]) == ["b", "d"]


def test_join_duplicate_col_error(small_pdf):
    a = AnovosFrame.from_pandas(small_pdf)
    b = AnovosFrame.from_pandas(small_pdf)
    with pytest.raises(ValueError):
        di.join_dataset(a, b, join_cols="ifa", join_type="inner")


def test_delete_select_rename_recast(ctx, small_pdf):
    f = AnovosFrame.from_pandas(small_pdf)
    assert "x" not in di.delete_column(f, ["x"]).columns
    assert di.select_column(f, "ifa|x").columns == ["ifa", "x"]
    assert "q" in di.rename_column(f, ["x"], ["q"]).columns
    rc = di.recast_column(f, ["x"], ["string"])
    assert rc.col("x").kind == "categorical"


def test_recommend_type(ctx):
    n = 500  # dynamic threshold 0.01*500 = 5 > 3 distinct values
    rng = np.random.default_rng(0)
    pdf = pd.DataFrame(
        {
            "low_card_num": rng.integers(0, 3, n).astype("float64"),
            "high_card_num": rng.normal(size=n),
            "numeric_as_str": [str(v) for v in rng.integers(0, 100000, n)],
            "real_cat": rng.choice(["a", "b"], n),
        }
    )
    f = AnovosFrame.from_pandas(pdf)
    odf = di.recommend_type(ctx, f)
    recs = dict(zip(odf["attribute"], odf["recommended_form"]))
    assert recs.get("low_card_num") == "categorical"
    assert recs.get("numeric_as_str") == "numerical"
    assert "high_card_num" not in recs
    assert "real_cat" not in recs
