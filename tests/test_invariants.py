"""Property-style invariant checks over randomized frames: whatever the
data, the analyzer's outputs must satisfy these algebraic facts."""

import numpy as np
import pandas as pd
import pytest
import torch

from anovos_amd.core.frame import AnovosFrame
from anovos_amd.data_analyzer import quality_checker as qc
from anovos_amd.data_analyzer import stats_generator as sg
from anovos_amd.drift_stability import drift_detector as dd
from anovos_amd.shared.context import init_context


@pytest.fixture
def ctx():
    return init_context("cpu")


def random_frame(seed, n=20_000):
    rng = np.random.default_rng(seed)
    pdf = pd.DataFrame(
        {
            "a": rng.normal(rng.uniform(-100, 100), rng.uniform(0.1, 50), n),
            "b": rng.lognormal(2, 1, n),
            "c": rng.integers(-5, 5, n).astype(float),
            "d": rng.choice([f"v{i}" for i in range(rng.integers(2, 30))], n),
        }
    )
    for col in ("a", "b", "c"):
        pdf.loc[rng.random(n) < rng.uniform(0, 0.3), col] = np.nan
    pdf.loc[rng.random(n) < 0.1, "d"] = None
    return AnovosFrame.from_pandas(pdf, device="cpu")


@pytest.mark.parametrize("seed", [1, 2, 3, 4])
def test_count_invariants(ctx, seed):
    idf = random_frame(seed)
    total = idf.count()
    m = sg.measures_of_counts(ctx, idf).set_index("attribute")
    for attr in m.index:
        assert int(m.loc[attr, "fill_count"]) + int(m.loc[attr, "missing_count"]) == total
        assert 0 <= float(m.loc[attr, "missing_pct"]) <= 1


@pytest.mark.parametrize("seed", [5, 6, 7])
def test_percentile_monotonicity(ctx, seed):
    idf = random_frame(seed, n=400_000)  # sketch path
    p = sg.measures_of_percentiles(ctx, idf, ["a", "b", "c"]).set_index("attribute")
    qs = ["1%", "5%", "10%", "25%", "50%", "75%", "90%", "95%", "99%"]
    for attr in p.index:
        vals = [float(p.loc[attr, q]) for q in qs]
        vals = [v for v in vals if v == v]
        assert vals == sorted(vals), (attr, vals)


@pytest.mark.parametrize("seed", [8, 9])
def test_dispersion_invariants(ctx, seed):
    idf = random_frame(seed)
    d = sg.measures_of_dispersion(ctx, idf).set_index("attribute")
    for attr in d.index:
        sd = float(d.loc[attr, "stddev"])
        var = float(d.loc[attr, "variance"])
        if sd == sd:
            assert sd >= 0
            assert var == pytest.approx(sd * sd, rel=5e-3)  # table rounds to 4 decimals
        iqr = float(d.loc[attr, "IQR"])
        if iqr == iqr:
            assert iqr >= -1e-9


@pytest.mark.parametrize("seed", [10, 11])
def test_drift_invariants(ctx, seed):
    idf = random_frame(seed)
    rng = np.random.default_rng(seed + 100)
    from anovos_amd.core.frame import Column

    shifted = idf.with_column("a", Column("a", "float", idf.col("a").data + rng.uniform(0, 5)))
    import tempfile

    with tempfile.TemporaryDirectory() as tmp:
        stats = dd.statistics(ctx, shifted, idf, list_of_cols=["a", "b"], method_type="all",
                              use_sampling=False, source_path=tmp)
    for _, r in stats.iterrows():
        assert float(r["PSI"]) >= -1e-9
        assert 0 <= float(r["HD"]) <= 1 + 1e-9
        assert float(r["JSD"]) >= -1e-9
        assert 0 <= float(r["KS"]) <= 1 + 1e-9
    # identical frames drift to zero
    with tempfile.TemporaryDirectory() as tmp:
        zero = dd.statistics(ctx, idf, idf, list_of_cols=["a", "b"], method_type="all",
                             use_sampling=False, source_path=tmp)
    assert (pd.to_numeric(zero["PSI"]) == 0).all()
    assert (zero["flagged"] == 0).all()


@pytest.mark.parametrize("seed", [12, 13])
def test_unique_counts_bounds(ctx, seed):
    idf = random_frame(seed, n=400_000)
    u = sg.uniqueCount_computation(ctx, idf, compute_approx_unique_count=True).set_index("attribute")
    m = sg.measures_of_counts(ctx, idf).set_index("attribute")
    for attr in u.index:
        uv = float(u.loc[attr, "unique_values"])
        fill = float(m.loc[attr, "fill_count"])
        assert 0 <= uv <= fill * 1.05 + 10  # HLL rsd margin


@pytest.mark.parametrize("seed", [14])
def test_nullrows_consistency(ctx, seed):
    idf = random_frame(seed)
    total = idf.count()
    _, nr = qc.nullRows_detection(ctx, idf, treatment=False)
    assert int(nr["row_count"].sum()) == total
    odf, _ = qc.duplicate_detection(ctx, idf, treatment=True)
    assert odf.count() <= total


@pytest.mark.parametrize("offset", [1e6, 1e8, 1e10, 1e12])
def test_moment_stability_large_offset(ctx, offset):
    """Skew/kurt survive |mean| >> stddev (shifted-pivot accumulation;
    raw fp64 power sums cancel catastrophically from offset ~1e5 on)."""
    from anovos_amd.core.frame import Column

    x = torch.randn(200_000, generator=torch.Generator().manual_seed(17), dtype=torch.float64) + offset
    idf = AnovosFrame({"x": Column("x", "double", x)}, device="cpu")
    d = sg.measures_of_dispersion(ctx, idf).set_index("attribute")
    s = sg.measures_of_shape(ctx, idf).set_index("attribute")
    assert float(d.loc["x", "stddev"]) == pytest.approx(1.0, rel=2e-2)
    assert abs(float(s.loc["x", "skewness"])) < 0.1
    assert abs(float(s.loc["x", "kurtosis"])) < 0.2


def test_correlation_large_offset(ctx):
    """Pearson/covariance survive |mean| >> spread (f64 centering before
    any f32 downcast)."""
    from anovos_amd.core.frame import Column
    from anovos_amd.ops import corr

    g = torch.Generator().manual_seed(3)
    n = 200_000
    z = torch.randn(n, generator=g, dtype=torch.float64)
    a = z + 1e9
    b = 0.5 * z + 0.866 * torch.randn(n, generator=g, dtype=torch.float64) + 1e9
    idf = AnovosFrame({"a": Column("a", "double", a), "b": Column("b", "double", b)}, device="cpu")
    assert corr.pearson_matrix(idf, ["a", "b"])[0, 1] == pytest.approx(0.5, abs=0.02)
    assert corr.covariance_matrix(idf, ["a", "b"])[0, 1] == pytest.approx(0.5, abs=0.02)


@pytest.mark.parametrize("seed", [21, 22])
def test_binning_invariants(ctx, seed):
    """Equal-range bin labels live in 1..bin_size (NaN for null); equal-
    frequency bins hold roughly n/bin_size rows each at the sketch's
    rank tolerance."""
    from anovos_amd.data_transformer import transformers as T

    idf = random_frame(seed, n=300_000)
    for method in ("equal_range", "equal_frequency"):
        odf = T.attribute_binning(ctx, idf, ["a", "b"], method_type=method,
                                  bin_size=10, output_mode="append")
        for c in ("a_binned", "b_binned"):
            v = odf.col(c).data
            vv = v[~torch.isnan(v)]
            assert float(vv.min()) >= 1 and float(vv.max()) <= 10
            if method == "equal_frequency":
                counts = torch.bincount(vv.long(), minlength=11)[1:]
                n = int(vv.numel())
                # each bucket within ~3x rank tolerance of n/10
                assert int(counts.max()) < n / 10 + 3 * 0.01 * n + 10


@pytest.mark.parametrize("seed", [23])
def test_z_standardization_roundtrip(ctx, seed):
    """Standardized columns have mean~0/sd~1 and invert back to the
    original values through the saved mean/sd."""
    from anovos_amd.data_analyzer import stats_generator as sg
    from anovos_amd.data_transformer import transformers as T

    idf = random_frame(seed)
    d0 = sg.measures_of_dispersion(ctx, idf, ["a"]).set_index("attribute")
    m0 = sg.measures_of_centralTendency(ctx, idf, ["a"]).set_index("attribute")
    odf = T.z_standardization(ctx, idf, ["a"], output_mode="append")
    z = odf.col("a_scaled").data
    zz = z[~torch.isnan(z)].to(torch.float64)
    assert abs(float(zz.mean())) < 1e-3
    assert float(zz.std(unbiased=True)) == pytest.approx(1.0, rel=1e-2)
    back = zz * float(d0.loc["a", "stddev"]) + float(m0.loc["a", "mean"])
    orig = idf.col("a").data
    oo = orig[~torch.isnan(orig)].to(torch.float64)
    assert float((back - oo).abs().max()) < 1e-3 * max(1.0, float(oo.abs().max()))


def test_duplicate_detection_idempotent(ctx):
    """Deduplicating twice equals deduplicating once."""
    import pandas as pd
    from anovos_amd.data_analyzer import quality_checker as qc

    rng = np.random.default_rng(31)
    pdf = pd.DataFrame({"a": rng.integers(0, 50, 5000).astype(float),
                        "b": rng.choice(["x", "y"], 5000)})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    once, _ = qc.duplicate_detection(ctx, idf, treatment=True)
    twice, stats2 = qc.duplicate_detection(ctx, once, treatment=True)
    assert once.count() == twice.count()
    assert once.count() == len(pdf.drop_duplicates())
