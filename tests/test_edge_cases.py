"""Edge-case hardening: degenerate columns through the full analyzer
surface (constant, all-null, single-value, ±inf, huge-cardinality).
Runs on CPU here; the same assertions run on GPU via test_gpu_workflow's
device-parametrized twin."""

import numpy as np
import pandas as pd
import pytest
import torch

from anovos_amd.core.frame import AnovosFrame, Column
from anovos_amd.data_analyzer import quality_checker as qc
from anovos_amd.data_analyzer import stats_generator as sg
from anovos_amd.data_transformer import transformers as T
from anovos_amd.shared.context import init_context


@pytest.fixture
def ctx():
    return init_context("cpu")


def edge_frame(device="cpu"):
    n = 5000
    g = torch.Generator().manual_seed(77)
    cols = {
        "normal": Column("normal", "float", torch.randn(n, generator=g)),
        "constant": Column("constant", "float", torch.full((n,), 3.14)),
        "all_null": Column("all_null", "float", torch.full((n,), float("nan"))),
        "with_inf": Column("with_inf", "float", torch.randn(n, generator=g)),
        "one_cat": Column("one_cat", "string", torch.zeros(n, dtype=torch.int32), ["only"]),
        "null_cat": Column("null_cat", "string", torch.full((n,), -1, dtype=torch.int32), []),
    }
    cols["with_inf"].data[::100] = float("inf")
    cols["with_inf"].data[50::100] = float("-inf")
    f = AnovosFrame(cols, device="cpu")
    return f.to_device(device) if device != "cpu" else f


def test_counts_and_modes_on_edges(ctx):
    idf = edge_frame()
    counts = sg.measures_of_counts(ctx, idf)
    m = counts.set_index("attribute")
    assert int(m.loc["all_null", "missing_count"]) == 5000
    assert int(m.loc["constant", "missing_count"]) == 0
    assert int(m.loc["null_cat", "missing_count"]) == 5000
    ct = sg.measures_of_centralTendency(ctx, idf)
    cm = ct.set_index("attribute")
    assert cm.loc["one_cat", "mode"] == "only"
    assert float(cm.loc["constant", "mean"]) == pytest.approx(3.14, rel=1e-6)


def test_dispersion_constant_and_null(ctx):
    idf = edge_frame()
    d = sg.measures_of_dispersion(ctx, idf).set_index("attribute")
    assert float(d.loc["constant", "stddev"]) == pytest.approx(0.0, abs=1e-9)
    # all-null column: stats are NaN, not crashes
    assert np.isnan(float(d.loc["all_null", "stddev"])) or d.loc["all_null", "stddev"] is None


def test_percentiles_and_binning_on_edges(ctx):
    idf = edge_frame()
    p = sg.measures_of_percentiles(ctx, idf, ["normal", "constant"]).set_index("attribute")
    assert float(p.loc["constant", "50%"]) == pytest.approx(3.14, rel=1e-6)
    # binning a constant column must not crash (degenerate range)
    odf = T.attribute_binning(ctx, idf, ["normal", "constant"], bin_size=5, output_mode="append")
    assert "constant_binned" in odf.columns


def test_inf_values_flow_through(ctx):
    idf = edge_frame()
    d = sg.measures_of_dispersion(ctx, idf, ["with_inf"])
    # inf contaminates the moments exactly like Spark's agg would;
    # the engine must not crash and min/max must reflect the infs
    shape = sg.measures_of_shape(ctx, idf, ["with_inf"])
    assert len(d) == 1 and len(shape) == 1
    mom = sg.global_summary(ctx, idf)
    assert mom is not None


def test_quality_checks_on_edges(ctx):
    idf = edge_frame()
    _, nr = qc.nullRows_detection(ctx, idf, treatment=False)
    assert int(nr["row_count"].sum()) == 5000  # every row has >= 1 null
    _, bias = qc.biasedness_detection(ctx, idf, ["one_cat", "null_cat"], treatment=False)
    b = bias.set_index("attribute")
    assert float(b.loc["one_cat", "mode_pct"]) == pytest.approx(1.0)
    odf, stats = qc.outlier_detection(ctx, idf, ["normal", "constant", "all_null"],
                                      detection_side="both", print_impact=True)
    assert len(stats) >= 1  # degenerate columns excluded or zero-flagged, no crash


def test_single_row_frame(ctx):
    idf = AnovosFrame({"x": Column("x", "float", torch.tensor([1.5]))}, device="cpu")
    c = sg.measures_of_counts(ctx, idf)
    assert int(c["fill_count"][0]) == 1
    d = sg.measures_of_dispersion(ctx, idf)
    assert len(d) == 1  # stddev NaN at n=1, no crash


def test_high_cardinality_int(ctx):
    n = 300_000
    g = torch.Generator().manual_seed(3)
    x = torch.randint(0, 2**31 - 1, (n,), generator=g).to(torch.float64)
    idf = AnovosFrame({"big": Column("big", "double", x)}, device="cpu")
    u = sg.uniqueCount_computation(ctx, idf, compute_approx_unique_count=True)
    est = float(u["unique_values"][0])
    assert abs(est - n) / n < 0.1  # nearly all distinct


def test_association_on_edges(ctx):
    """IV/IG with an all-null column, correlation with a constant column,
    and a label with a single event (round-2 fused K9 paths on CPU)."""
    import numpy as np
    import pandas as pd
    import torch

    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.data_analyzer import association_evaluator as ae

    n = 5000
    rng = np.random.default_rng(9)
    pdf = pd.DataFrame(
        {
            "x": rng.normal(0, 1, n),
            "const": np.ones(n),
            "allnull": np.full(n, np.nan),
            "cat": rng.choice(["a", "b"], n),
            "label": rng.choice(["0", "1"], n, p=[0.6, 0.4]),
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    iv = ae.IV_calculation(ctx, idf, label_col="label", event_label="1")
    assert set(iv["attribute"]) == {"x", "const", "allnull", "cat"}
    assert np.isfinite(iv["iv"]).all()
    ig = ae.IG_calculation(ctx, idf, label_col="label", event_label="1")
    assert np.isfinite(ig["ig"]).all()
    corr = ae.correlation_matrix(ctx, idf, ["x", "const", "allnull"])
    m = corr[sorted(["x", "const", "allnull"])].to_numpy()
    assert np.diag(m).tolist() == [1.0, 1.0, 1.0]  # diag forced to 1 even for degenerate cols

    # single-event label: IV still computes (0.5-smoothed WOE fallback)
    pdf2 = pdf.copy()
    lab = np.array(["0"] * n, dtype=object)
    lab[0] = "1"
    pdf2["label"] = lab
    idf2 = AnovosFrame.from_pandas(pdf2, device="cpu")
    iv2 = ae.IV_calculation(ctx, idf2, label_col="label", event_label="1")
    assert np.isfinite(iv2["iv"]).all()


def test_drift_with_missing_target_column(ctx):
    """Source has a column absent in the target: excluded with a warning,
    remaining metrics intact (workflow parity)."""
    import warnings as _w

    import numpy as np
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.drift_stability import drift_detector as dd

    rng = np.random.default_rng(5)
    n = 20000
    tgt = pd.DataFrame({"x": rng.normal(0, 1, n), "y": rng.normal(3, 1, n)})
    src = pd.DataFrame({"x": rng.normal(0.3, 1, n), "y": rng.normal(3, 1, n), "extra": rng.normal(0, 1, n)})
    with _w.catch_warnings():
        _w.simplefilter("ignore")
        stats = dd.statistics(ctx, AnovosFrame.from_pandas(tgt), AnovosFrame.from_pandas(src),
                              list_of_cols="all", method_type="all", use_sampling=False,
                              model_directory="/tmp/_drift_edge")
    assert set(stats["attribute"]) <= {"x", "y"}
    row = stats[stats["attribute"] == "x"].iloc[0]
    assert float(row["PSI"]) > 0


def test_stability_two_snapshots(ctx):
    import numpy as np
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.drift_stability import stability as st

    rng = np.random.default_rng(3)
    idfs = [AnovosFrame.from_pandas(pd.DataFrame({"v": rng.normal(10, 1 + 0.2 * k, 5000)})) for k in range(2)]
    out = st.stability_index_computation(ctx, *idfs)
    assert "stability_index" in out.columns and len(out) == 1
    assert 0 <= float(out["stability_index"].iloc[0]) <= 4
