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


def test_io_roundtrip_all_formats_and_dtypes(tmp_path):
    """Property: write -> read across csv/parquet/json/avro preserves
    values, dtype KINDS and dictionary contents — exercises the r02
    Arrow-native ingest (dictionary_encode fast path, timestamp cast,
    nullable ints) against the pandas fallback semantics."""
    import numpy as np
    import pandas as pd
    import torch

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.core.io import read_dataset, write_dataset

    rng = np.random.default_rng(13)
    n = 5000
    pdf = pd.DataFrame(
        {
            "f32like": rng.normal(0, 1, n).astype("float32"),
            "f64": rng.normal(1e9, 1, n),
            "int_clean": rng.integers(-50, 50, n),
            "big": rng.integers(0, 2**40, n),
            "s": rng.choice(["alpha", "beta", "gamma", ""], n).astype(object),
            "ts": pd.to_datetime(rng.integers(1_500_000_000, 1_700_000_000, n), unit="s"),
            "b": rng.random(n) < 0.5,
        }
    )
    pdf.loc[rng.choice(n, 100, replace=False), "s"] = None
    pdf.loc[rng.choice(n, 100, replace=False), "f64"] = np.nan
    src = AnovosFrame.from_pandas(pdf, device="cpu")

    for fmt in ("parquet", "csv", "json", "avro"):
        d = str(tmp_path / fmt)
        write_dataset(src, d, fmt, {"mode": "overwrite", "header": True})
        back = read_dataset(d, fmt, {"header": True, "inferSchema": True})
        assert back.local_rows() == n, fmt
        for c in src.columns:
            a, b = src.col(c), back.col(c)
            assert a.kind == b.kind, (fmt, c, a.kind, b.kind)
            if a.kind == "numerical":
                av = a.data.to(torch.float64)
                bv = b.data.to(torch.float64)
                both = ~(torch.isnan(av) | torch.isnan(bv))
                assert torch.isnan(av).equal(torch.isnan(bv)), (fmt, c)
                tol = 1e-4 if fmt in ("csv", "json") else 1e-6  # text formats round-trip via decimal
                assert torch.allclose(av[both], bv[both], rtol=tol, atol=tol), (fmt, c)
            elif a.kind == "categorical":
                ao = a.to_numpy_objects()
                bo = b.to_numpy_objects()
                # "" and null may merge in text formats; compare non-null equality
                for x, y in zip(ao.tolist()[:500], bo.tolist()[:500]):
                    if x is None or x == "":
                        assert y is None or y == "", (fmt, c, x, y)
                    else:
                        assert x == y, (fmt, c, x, y)


def test_write_dataset_negative_repartition_raises(tmp_path, small_pdf):
    """Reference test_data_ingest_unit.py:79 — Spark raised
    IllegalArgumentException; the engine rejects the same input."""
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_ingest.data_ingest import write_dataset

    idf = AnovosFrame.from_pandas(small_pdf, device="cpu")
    with pytest.raises(ValueError):
        write_dataset(idf, str(tmp_path / "out"), "csv",
                      {"header": True, "delimiter": ",", "repartition": -2})


def test_write_dataset_column_order(tmp_path, small_pdf):
    """Reference test_data_ingest_unit.py:97/:119 — column_order is
    honored in the written file; a wrong-length order raises."""
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_ingest.data_ingest import write_dataset

    idf = AnovosFrame.from_pandas(small_pdf, device="cpu")
    order = list(reversed(idf.columns))
    write_dataset(idf, str(tmp_path / "ordered"), "csv", {"header": True, "mode": "overwrite"},
                  column_order=order)
    import glob

    part = sorted(glob.glob(str(tmp_path / "ordered" / "part-*")))[0]
    assert list(pd.read_csv(part).columns) == order
    with pytest.raises(ValueError):
        write_dataset(idf, str(tmp_path / "bad"), "csv", {"header": True, "mode": "overwrite"},
                      column_order=order[:-1])
