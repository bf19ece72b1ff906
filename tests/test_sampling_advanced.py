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
