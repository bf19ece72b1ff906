import numpy as np
import pandas as pd
import pytest

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_analyzer import association_evaluator as ae


@pytest.fixture(scope="module")
def assoc_frame():
    rng = np.random.default_rng(21)
    n = 3000
    x = rng.normal(0, 1, n)
    y = 0.9 * x + rng.normal(0, 0.3, n)  # strongly correlated with x
    z = rng.normal(0, 1, n)  # independent
    w = 0.9 * z + rng.normal(0, 0.3, n)  # correlated with z
    score = x + rng.normal(0, 0.5, n)
    label = np.where(score > 0.2, ">50K", "<=50K")
    cat = rng.choice(["a", "b", "c"], n)
    return AnovosFrame.from_pandas(pd.DataFrame({"x": x, "y": y, "z": z, "w": w, "cat": cat, "label": label}))


def test_correlation_matrix(ctx, assoc_frame):
    odf = ae.correlation_matrix(ctx, assoc_frame, ["x", "y", "z"]).set_index("attribute")
    assert abs(odf.loc["x", "x"] - 1.0) < 1e-6
    assert odf.loc["x", "y"] > 0.9
    assert abs(odf.loc["x", "z"]) < 0.1
    # numpy reference check
    pdf = assoc_frame.to_pandas()[["x", "y", "z"]].astype(float)
    expected = pdf.corr().loc["x", "y"]
    assert abs(odf.loc["x", "y"] - expected) < 1e-4


def test_IV_calculation(ctx, assoc_frame):
    odf = ae.IV_calculation(ctx, assoc_frame, label_col="label", event_label=">50K").set_index("attribute")
    assert odf.loc["x", "iv"] > odf.loc["z", "iv"]
    assert odf.loc["x", "iv"] > 0.5
    assert odf.loc["cat", "iv"] < 0.1


def test_IG_calculation(ctx, assoc_frame):
    odf = ae.IG_calculation(ctx, assoc_frame, label_col="label", event_label=">50K").set_index("attribute")
    assert odf.loc["x", "ig"] > odf.loc["z", "ig"]
    assert odf.loc["x", "ig"] > 0.1


def test_variable_clustering(ctx, assoc_frame):
    odf = ae.variable_clustering(ctx, assoc_frame, ["x", "y", "z", "w"])
    assert set(odf.columns) == {"Cluster", "Attribute", "RS_Ratio"}
    cl = dict(zip(odf["Attribute"], odf["Cluster"]))
    assert cl["x"] == cl["y"]  # correlated pair clusters together
    assert cl["z"] == cl["w"]
    assert cl["z"] != cl["x"]


def test_quartimax_rotation():
    from anovos_amd.data_analyzer.association_eval_varclus import quartimax_rotation

    rng = np.random.default_rng(0)
    L = rng.normal(size=(6, 2))
    R = quartimax_rotation(L)
    # rotation preserves column space / frobenius norm
    assert abs(np.linalg.norm(R) - np.linalg.norm(L)) < 1e-8
    assert np.sum(R**4) >= np.sum(L**4) - 1e-9  # criterion not decreased


def test_correlation_matrix_algebraic_invariants(ctx, assoc_frame):
    """Symmetric, unit diagonal, entries in [-1,1], positive
    semi-definite up to numerical tolerance."""
    out = ae.correlation_matrix(ctx, assoc_frame, list_of_cols=["x", "y", "z", "w"])
    m = out.set_index("attribute").loc[["x", "y", "z", "w"], ["x", "y", "z", "w"]].to_numpy(dtype=float)
    assert np.allclose(m, m.T, atol=1e-9)
    assert np.allclose(np.diag(m), 1.0)
    assert (np.abs(m) <= 1 + 1e-9).all()
    assert np.linalg.eigvalsh(m).min() > -1e-6


def test_iv_separability_ordering(ctx, assoc_frame):
    """The label-driving attribute must dominate IV and IG; an
    independent attribute must land near zero (IV < 0.1 = weak per the
    reference's own interpretation bands)."""
    iv = ae.IV_calculation(ctx, assoc_frame, list_of_cols=["x", "z"], label_col="label", event_label=">50K").set_index("attribute")
    ig = ae.IG_calculation(ctx, assoc_frame, list_of_cols=["x", "z"], label_col="label", event_label=">50K").set_index("attribute")
    assert float(iv.loc["x", "iv"]) > 10 * max(float(iv.loc["z", "iv"]), 1e-6)
    assert float(iv.loc["z", "iv"]) < 0.1
    assert float(ig.loc["x", "ig"]) > float(ig.loc["z", "ig"])
