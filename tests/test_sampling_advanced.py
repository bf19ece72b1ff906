"""data_sampling + advanced transformer tests (reference parity:
test_data_sampling.py 2 tests; transformers advanced paths)."""

import numpy as np
import pandas as pd
import pytest
import torch

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_ingest.data_sampling import data_sample
from anovos_amd.data_transformer import transformers_advanced as TA
from anovos_amd.shared.context import init_context


@pytest.fixture
def ctx():
    return init_context("cpu")


@pytest.fixture
def frame():
    rng = np.random.default_rng(3)
    n = 5000
    pdf = pd.DataFrame(
        {
            "x": rng.normal(10, 2, n),
            "y": rng.normal(-5, 1, n),
            "z": rng.normal(0, 3, n),
            "strata": rng.choice(["a", "b", "c"], n, p=[0.6, 0.3, 0.1]),
        }
    )
    pdf.loc[rng.choice(n, 200, replace=False), "x"] = np.nan
    return AnovosFrame.from_pandas(pdf, device="cpu")


def test_random_sampling(ctx, frame):
    odf = data_sample(frame, strata_cols="all", drop_cols=[], fraction=0.2, method_type="random")
    n = odf.local_rows()
    assert 700 <= n <= 1300  # ~1000 expected


def test_stratified_population_sampling(ctx, frame):
    odf = data_sample(frame, strata_cols=["strata"], fraction=0.3,
                      method_type="stratified", stratified_type="population")
    pdf = odf.to_pandas()
    shares = pdf["strata"].value_counts(normalize=True)
    # proportionate: shares close to 0.6/0.3/0.1
    assert abs(shares.get("a", 0) - 0.6) < 0.1
    assert abs(shares.get("c", 0) - 0.1) < 0.06


def test_stratified_balanced_sampling(ctx, frame):
    odf = data_sample(frame, strata_cols=["strata"], fraction=0.5,
                      method_type="stratified", stratified_type="balanced")
    pdf = odf.to_pandas()
    counts = pdf["strata"].value_counts()
    # balanced: all strata approximately equal counts
    assert counts.max() - counts.min() <= max(40, counts.max() * 0.25)


def test_imputation_sklearn_knn(ctx, frame):
    odf = TA.imputation_sklearn(ctx, frame, list_of_cols=["x", "y", "z"], method_type="KNN",
                                sample_size=2000)
    assert int(odf.col("x").null_mask().sum()) == 0


def test_imputation_matrix_factorization(ctx, frame):
    odf = TA.imputation_matrixFactorization(ctx, frame, list_of_cols=["x", "y", "z"], id_col="")
    assert int(odf.col("x").null_mask().sum()) == 0


def test_auto_imputation_selects_a_method(ctx, frame):
    odf = TA.auto_imputation(ctx, frame, list_of_cols=["x", "y", "z"], null_pct=0.1)
    assert int(odf.col("x").null_mask().sum()) == 0


def test_pca_latent_features(ctx, frame):
    odf = TA.PCA_latentFeatures(ctx, frame, list_of_cols=["x", "y", "z"],
                                explained_variance_cutoff=0.6, output_mode="append")
    latents = [c for c in odf.columns if c.startswith("latent_")]
    assert len(latents) >= 1
    assert odf.col(latents[0]).data.shape[0] == frame.local_rows()


def test_autoencoder_latent_features(ctx, frame):
    odf = TA.autoencoder_latentFeatures(ctx, frame, list_of_cols=["x", "y", "z"],
                                        reduction_params=0.5, epochs=3, sample_size=1000,
                                        output_mode="append")
    latents = [c for c in odf.columns if c.startswith("latent_")]
    assert len(latents) >= 1


def test_imputation_sklearn_device_apply_matches_sklearn():
    """The tensorized KNN / IterativeImputer apply (K15) must reproduce
    sklearn's own .transform on the same fitted model (the round-1 host
    loop was the reference; now it IS the device path run on CPU)."""
    import numpy as np
    import torch
    from sklearn.experimental import enable_iterative_imputer  # noqa: F401
    from sklearn.impute import IterativeImputer, KNNImputer

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_transformer import transformers_advanced as TA
    from anovos_amd.shared.context import init_context

    rng = np.random.default_rng(3)
    n = 4000
    pdf = pd.DataFrame(
        {
            "a": rng.normal(10, 3, n),
            "b": rng.normal(-5, 2, n),
            "c": rng.normal(0, 1, n),
            "d": rng.normal(100, 20, n),
        }
    )
    pdf["b"] = pdf["a"] * 0.5 + rng.normal(0, 0.3, n)  # correlated for regression
    for c in pdf.columns:
        pdf.loc[rng.choice(n, 160, replace=False), c] = np.nan
    ctx = init_context("cpu")
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    cols = list(pdf.columns)

    for method, klass in (("KNN", KNNImputer), ("regression", IterativeImputer)):
        out = TA.imputation_sklearn(ctx, idf, cols, method_type=method,
                                    use_sampling=True, sample_size=1500, sample_seed=42)
        # rebuild the same fitted model (same sample path) for the truth
        got = np.column_stack([out.col(c).data.numpy() for c in cols])
        X = pdf.to_numpy()
        assert not np.isnan(got).any()
        # fit on the same sampled data the engine used: easiest faithful
        # check — refit on FULL data and compare against sklearn's own
        # transform of the engine's model is internal; instead verify
        # against sklearn transform with an identical fit sample
        from anovos_amd.data_ingest.data_sampling import data_sample

        sub = data_sample(idf.select(cols), strata_cols="all", fraction=1500 / n,
                          method_type="random", stratified_type="population", seed_value=42)
        local = np.column_stack([sub.col(c).data.numpy().astype("float64") for c in cols])
        model = klass(n_neighbors=5) if method == "KNN" else klass(max_iter=10, random_state=42)
        model.fit(local)
        truth = model.transform(X)
        assert np.allclose(got, truth.astype(np.float32), rtol=2e-4, atol=2e-3), (
            method, np.nanmax(np.abs(got - truth)))


def test_matrix_factorization_batched_solver():
    """Row-batched ALS (no n*k*k materialization) still reconstructs a
    low-rank matrix's missing entries."""
    import numpy as np
    import torch

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_transformer import transformers_advanced as TA
    from anovos_amd.shared.context import init_context

    rng = np.random.default_rng(5)
    n, m, k = 3000, 8, 3
    U = rng.normal(0, 1, (n, k))
    V = rng.normal(0, 1, (m, k))
    X = U @ V.T
    pdf = pd.DataFrame(X, columns=[f"c{j}" for j in range(m)])
    mask = rng.random((n, m)) < 0.15
    truth = pdf.copy()
    pdf = pdf.mask(mask)
    ctx = init_context("cpu")
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    out = TA.imputation_matrixFactorization(ctx, idf, "all", rank=5, max_iter=15)
    err_num, err_den = 0.0, 0.0
    for j in range(m):
        got = out.col(f"c{j}").data.numpy()
        sel = mask[:, j]
        err_num += float(np.sum((got[sel] - truth.iloc[:, j].to_numpy()[sel]) ** 2))
        err_den += float(np.sum(truth.iloc[:, j].to_numpy()[sel] ** 2))
    assert err_num / max(err_den, 1e-9) < 0.05  # <5% relative MSE on held-out entries


def test_imputation_matrixFactorization_empty_cols_noop(ctx, frame):
    """Reference test_transformers.py:383 — empty list_of_cols returns
    the input unchanged."""
    odf = TA.imputation_matrixFactorization(ctx, frame, list_of_cols=[], id_col="")
    assert odf.columns == frame.columns
    assert torch.equal(torch.nan_to_num(odf.col("x").data), torch.nan_to_num(frame.col("x").data))


def test_imputation_matrixFactorization_append(ctx, frame):
    """Reference test_transformers.py:397 — append mode adds *_imputed
    columns and keeps the originals untouched."""
    odf = TA.imputation_matrixFactorization(ctx, frame, list_of_cols=["x", "y"], id_col="",
                                             output_mode="append")
    assert "x_imputed" in odf.columns and "y_imputed" in odf.columns
    assert torch.equal(torch.nan_to_num(odf.col("x").data), torch.nan_to_num(frame.col("x").data))
    assert not bool(torch.isnan(odf.col("x_imputed").data).any())


def test_auto_imputation_empty_cols_noop(ctx, frame):
    """Reference test_transformers.py:485."""
    odf = TA.auto_imputation(ctx, frame, list_of_cols=[], null_pct=0.1)
    odf = odf[0] if isinstance(odf, tuple) else odf
    assert odf.columns == frame.columns
