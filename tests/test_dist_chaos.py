"""Pathological-shard battery at world_size=2 (gloo, CPU).

Runs the full analyzer surface (7 stats-generator metrics, 7 quality
checkers, IV/IG, correlation) on a frame with every nasty column shape
— all-null numeric/categorical, constant, integral-float, per-row-
unique strings, interleaved nulls — sharded three ways (even, rank-1
EMPTY, rank-1 3 rows), and requires rank-0's results to EQUAL the
single-process truth. This is the generic guard for the class of
rank-divergence bugs fixed in r02 (local decisions feeding collective
counts): any new local-data-dependent branch in a merge path either
deadlocks (join timeout -> nonzero exit) or miscounts (mismatch) here.
"""

import json
import multiprocessing as mp
import os
import socket

import numpy as np
import pandas as pd
import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _make_nasty(n=500, seed=0):
    rng = np.random.default_rng(seed)
    df = pd.DataFrame({
        "num": rng.normal(0, 1, n),
        "allnull": [np.nan] * n,
        "const": [7.0] * n,
        "intlike": rng.integers(0, 5, n).astype(float),
        "cat": rng.choice(["a", "b", "c"], n),
        "allnullcat": [None] * n,
        "constcat": ["only"] * n,
        "unique_str": [f"u{i:05d}" for i in range(n)],
        "label": rng.choice(["0", "1"], n),
    })
    df.loc[df.index % 7 == 0, "num"] = np.nan
    df.loc[df.index % 5 == 0, "cat"] = None
    return df


def _battery(ctx, idf):
    from anovos_amd.data_analyzer import association_evaluator as ae
    from anovos_amd.data_analyzer import quality_checker as qc
    from anovos_amd.data_analyzer import stats_generator as sg

    out = {}
    for f in [sg.global_summary, sg.measures_of_counts, sg.measures_of_centralTendency,
              sg.measures_of_cardinality, sg.measures_of_dispersion,
              sg.measures_of_percentiles, sg.measures_of_shape]:
        r = f(ctx, idf, drop_cols=["label"])
        out[f.__name__] = r.fillna(-999).astype(str).to_dict("records")
    for f in [qc.duplicate_detection, qc.nullRows_detection]:
        r = f(ctx, idf)
        r = r[1] if isinstance(r, tuple) else r
        out[f.__name__] = r.fillna(-999).astype(str).to_dict("records")
    for f in [qc.nullColumns_detection, qc.outlier_detection, qc.IDness_detection,
              qc.biasedness_detection, qc.invalidEntries_detection]:
        r = f(ctx, idf, drop_cols=["label"])
        r = r[1] if isinstance(r, tuple) else r
        out[f.__name__] = r.fillna(-999).astype(str).to_dict("records")
    out["IV"] = ae.IV_calculation(ctx, idf, label_col="label", event_label="1").fillna(-999).astype(str).to_dict("records")
    out["IG"] = ae.IG_calculation(ctx, idf, label_col="label", event_label="1").fillna(-999).astype(str).to_dict("records")
    out["corr"] = ae.correlation_matrix(ctx, idf, drop_cols=["label"]).round(6).fillna(-999).astype(str).to_dict("records")
    vc = ae.variable_clustering(ctx, idf, drop_cols=["label", "allnull", "allnullcat", "unique_str"])
    out["varclus"] = vc.fillna(-999).astype(str).to_dict("records")

    # transformers: MMM null treatment + binning + z-scaling must give the
    # same treated values from global (not per-shard) statistics
    import torch

    from anovos_amd.core import dist as _dist
    from anovos_amd.data_transformer import transformers as tf

    # treated values must come from GLOBAL statistics: checks are local
    # partials merged through the same collectives the engine uses (a
    # no-op single-process), so dist == truth iff the fitted params match
    t1 = tf.imputation_MMM(ctx, idf, list_of_cols=["num", "intlike", "cat"])
    out["mmm_num_sum"] = round(_dist.all_reduce_scalar(
        float(torch.nansum(t1.col("num").data.to(torch.float64)))), 6)
    t2 = tf.attribute_binning(ctx, idf, list_of_cols=["num"], method_type="equal_frequency",
                              bin_size=4, output_mode="append")
    b = t2.col("num_binned").data
    cnts = torch.bincount(torch.nan_to_num(b, nan=0.0).to(torch.long), minlength=6)
    out["bin_counts"] = [int(v) for v in _dist.all_reduce_scalars([float(c) for c in cnts])]
    t3 = tf.z_standardization(ctx, idf, list_of_cols=["num"], output_mode="append")
    z = t3.col("num_zscaled" if "num_zscaled" in t3.columns else "num_scaled").data
    out["z_mean"] = round(_dist.all_reduce_scalar(float(torch.nansum(z.to(torch.float64)))), 4)
    return out


def _chaos_worker(rank, port, split, out_path):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank),
                       "WORLD_SIZE": str(len(split)),
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
                       "ANOVOS_AMD_DIST_BACKEND": "gloo"})
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    ctx = init_context("cpu")
    pdf = _make_nasty()
    lo, hi = split[rank]
    idf = AnovosFrame.from_pandas(pdf.iloc[lo:hi].reset_index(drop=True), device="cpu")
    res = _battery(ctx, idf)
    if rank == 0:
        json.dump(res, open(out_path, "w"))
    td.barrier()
    td.destroy_process_group()


@pytest.mark.parametrize("split", [
    pytest.param([(0, 250), (250, 500)], id="even"),
    pytest.param([(0, 500), (500, 500)], id="empty_rank1"),
    pytest.param([(0, 497), (497, 500)], id="tiny_rank1"),
    pytest.param([(0, 200), (200, 201), (201, 500)], id="three_ranks_skewed"),
])
def test_dist_chaos_battery(tmp_path, split):
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.shared.context import init_context

    ctx = init_context("cpu")
    truth = _battery(ctx, AnovosFrame.from_pandas(_make_nasty(), device="cpu"))

    port = _free_port()
    out = str(tmp_path / "res.json")
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_chaos_worker, args=(r, port, split, out)) for r in range(len(split))]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    res = json.load(open(out))
    bad = [k for k in truth if res.get(k) != truth[k]]
    assert not bad, f"distributed != single-process for: {bad}"
