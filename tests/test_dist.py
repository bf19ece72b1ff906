"""Multi-process distributed correctness (gloo, world_size=2 on CPU).

The reference had no distributed tests (Spark semantics assumed —
SURVEY.md §4); the rebuild must verify its RCCL partial-aggregate merge
protocol explicitly. Each test spawns 2 processes over gloo, shards a
frame by rank, runs the distributed op, and compares rank-0's result to
the single-process result on the unsharded frame."""

import json
import multiprocessing as mp
import os
import socket
import sys
import tempfile

import numpy as np
import pandas as pd
import pytest
import torch


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    p = s.getsockname()[1]
    s.close()
    return p


def _make_pdf(n=4000, seed=0):
    rng = np.random.default_rng(seed)
    return pd.DataFrame(
        {
            "x": rng.normal(5, 2, n),
            "y": rng.lognormal(1, 0.5, n),
            "cat": rng.choice(["a", "b", "c", "d"], n, p=[0.4, 0.3, 0.2, 0.1]),
            "label": rng.choice(["0", "1"], n, p=[0.7, 0.3]),
        }
    )


def _worker(rank, world, port, fn_name, out_path):
    os.environ.update(
        {
            "RANK": str(rank),
            "LOCAL_RANK": str(rank),
            "WORLD_SIZE": str(world),
            "MASTER_ADDR": "127.0.0.1",
            "MASTER_PORT": str(port),
        }
    )
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    ctx = init_context("cpu")
    pdf = _make_pdf()
    half = len(pdf) // 2
    shard = pdf.iloc[rank * half : (rank + 1) * half]
    idf = AnovosFrame.from_pandas(shard.reset_index(drop=True), device="cpu")

    result = _FNS[fn_name](ctx, idf)
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump(result, f)
    td.barrier()
    td.destroy_process_group()


def _fn_count_moments(ctx, idf):
    from anovos_amd.data_analyzer import stats_generator as sg

    disp = sg.measures_of_dispersion(ctx, idf)
    counts = sg.measures_of_counts(ctx, idf)
    return {
        "rows": idf.count(),
        "stddev_x": float(disp[disp["attribute"] == "x"]["stddev"].iloc[0]),
        "mean_fill": float(counts[counts["attribute"] == "x"]["fill_count"].iloc[0]),
    }


def _fn_quantiles_mode(ctx, idf):
    from anovos_amd.data_analyzer import stats_generator as sg

    pct = sg.measures_of_percentiles(ctx, idf)
    ct = sg.measures_of_centralTendency(ctx, idf)
    return {
        "p50_x": float(pct[pct["attribute"] == "x"]["50%"].iloc[0]),
        "mode_cat": str(ct[ct["attribute"] == "cat"]["mode"].iloc[0]),
        "unique_cat": float(sg.uniqueCount_computation(ctx, idf)["unique_values"].iloc[2]),
    }


def _fn_drift(ctx, idf):
    from anovos_amd.drift_stability import drift_detector as dd

    # target = slightly shifted copy of the local shard
    shifted = idf.copy()
    from anovos_amd.core.frame import Column

    x = shifted.col("x").data + 0.5
    shifted = shifted.with_column("x", Column("x", "float", x))
    # model_path must be SHARED across ranks (rank 0 writes, all read)
    shared = os.path.join(tempfile.gettempdir(), f"anovos_dist_drift_{os.environ.get('MASTER_PORT', '0')}")
    stats = dd.statistics(ctx, shifted, idf, list_of_cols=["x", "y"], method_type="all",
                          use_sampling=False, model_directory=shared + "/drift")
    row = stats[stats["attribute"] == "x"].iloc[0]
    return {"psi_x": float(row["PSI"]), "flagged": int(pd.to_numeric(stats["flagged"]).sum())}


def _fn_transforms(ctx, idf):
    import torch as _t

    from anovos_amd.core import dist as _dist
    from anovos_amd.data_transformer import transformers as T

    enc = T.cat_to_num_unsupervised(ctx, idf, ["cat"], method_type="label_encoding", output_mode="append")
    oc = T.outlier_categories(ctx, idf, ["cat"], max_category=3, coverage=1.0, output_mode="append")
    z = T.z_standardization(ctx, idf, ["x"], output_mode="append")
    # global artifacts must be rank-identical: kept categories + the
    # globally-standardized sum (≈0 when mean/stddev were global)
    kept = sorted(set(oc.col("cat_outliered").dictionary))
    zsum = float(_dist.all_reduce_scalar(float(z.col("x_scaled").data.to(_t.float64).sum())))
    enc_mean = float(_dist.all_reduce_scalar(float(enc.col("cat_index").data.to(_t.float64).sum()))) / max(idf.count(), 1)
    return {"kept_categories": kept, "zsum_near_zero": abs(zsum) < 1.0, "enc_mean": enc_mean}


_FNS = {
    "count_moments": _fn_count_moments,
    "quantiles_mode": _fn_quantiles_mode,
    "drift": _fn_drift,
    "transforms": _fn_transforms,
}


def _run_dist(fn_name):
    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name
    procs = []
    mp_ctx = mp.get_context("spawn")
    for r in range(2):
        p = mp_ctx.Process(target=_worker, args=(r, 2, port, fn_name, out))
        p.start()
        procs.append(p)
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    with open(out) as f:
        return json.load(f)


def _run_single(fn_name):
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.shared.context import init_context

    ctx = init_context("cpu")
    idf = AnovosFrame.from_pandas(_make_pdf(), device="cpu")
    return _FNS[fn_name](ctx, idf)


@pytest.mark.parametrize("fn_name", ["count_moments", "quantiles_mode", "drift", "transforms"])
def test_dist_matches_single(fn_name):
    dist_res = _run_dist(fn_name)
    single = _run_single(fn_name)
    for k, v in single.items():
        if isinstance(v, (int, float)):
            assert dist_res[k] == pytest.approx(v, rel=2e-2, abs=1e-6), (k, dist_res[k], v)
        else:
            assert dist_res[k] == v, (k, dist_res[k], v)


def _fn_skewed_dicts(ctx, idf):
    """Rank-disjoint categories must still merge correctly (dictionary
    unify at ingest)."""
    from anovos_amd.data_analyzer import stats_generator as sg

    ct = sg.measures_of_centralTendency(ctx, idf, ["skewcat"])
    u = sg.uniqueCount_computation(ctx, idf, ["skewcat"])
    return {"mode": str(ct["mode"].iloc[0]), "uniques": float(u["unique_values"].iloc[0])}


_FNS["skewed_dicts"] = _fn_skewed_dicts


def test_dist_rank_disjoint_dictionaries(tmp_path):
    """Write a dataset whose category values cluster by file part (so
    each rank's local dictionary differs), read it distributed, and
    check mode/uniques match the single-process truth."""
    import pandas as pd

    d = tmp_path / "skew"
    (d).mkdir()
    # part 0: categories a,a,a,b ; part 1: categories c,c,b,d
    pd.DataFrame({"skewcat": ["a", "a", "a", "b"] * 200}).to_csv(d / "part-00000.csv", index=False)
    pd.DataFrame({"skewcat": ["c", "c", "b", "d"] * 200}).to_csv(d / "part-00001.csv", index=False)

    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name

    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_dict_worker, args=(r, port, str(d), out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    res = json.load(open(out))
    # truth over all 1600 rows: counts a=600, b=400, c=400, d=200 -> mode a, 4 uniques
    assert res["mode"] == "a"
    assert res["uniques"] == 4.0


def _dict_worker(rank, port, data_dir, out):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.io import read_dataset
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    ctx = init_context("cpu")
    idf = read_dataset(data_dir, "csv", {"header": True})
    res = _fn_skewed_dicts(ctx, idf)
    if rank == 0:
        with open(out, "w") as f:
            json.dump(res, f)
    td.barrier()
    td.destroy_process_group()


def _join_worker(rank, port, data_dir, out):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.core.io import read_dataset
    from anovos_amd.data_ingest.data_ingest import join_dataset
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    init_context("cpu")
    main = read_dataset(os.path.join(data_dir, "main"), "csv", {"header": True})
    side = read_dataset(os.path.join(data_dir, "side"), "csv", {"header": True})
    joined = join_dataset(main, side, join_cols="k", join_type="inner")
    local = int(joined.local_rows())
    tot = dist.all_reduce_scalar(local)
    vsum = dist.all_reduce_scalar(float(joined.col("v").data.to(torch.float64).sum()))
    if rank == 0:
        with open(out, "w") as f:
            json.dump({"rows": tot, "vsum": vsum}, f)
    td.barrier()
    td.destroy_process_group()


def test_dist_join_sharded_side_table(tmp_path):
    """The side table's parts land on different ranks; the broadcast
    join must still match every key (reference join semantics)."""
    import pandas as pd

    d = tmp_path / "jd"
    (d / "main").mkdir(parents=True)
    (d / "side").mkdir()
    main = pd.DataFrame({"k": [f"k{i % 50}" for i in range(1000)], "x": range(1000)})
    main.iloc[:500].to_csv(d / "main" / "part-00000.csv", index=False)
    main.iloc[500:].to_csv(d / "main" / "part-00001.csv", index=False)
    side = pd.DataFrame({"k": [f"k{i}" for i in range(50)], "v": [float(i) for i in range(50)]})
    side.iloc[:25].to_csv(d / "side" / "part-00000.csv", index=False)
    side.iloc[25:].to_csv(d / "side" / "part-00001.csv", index=False)

    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_join_worker, args=(r, port, str(d), out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    res = json.load(open(out))
    assert res["rows"] == 1000  # every main row matches
    truth = sum(float(i % 50) for i in range(1000))
    assert res["vsum"] == pytest.approx(truth)


def _dedup_worker(rank, port, out):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_analyzer.quality_checker import duplicate_detection
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    ctx = init_context("cpu")
    # rank 0 holds values 0..99 (each twice); rank 1 holds 50..149 (each
    # twice): cross-rank dupes for 50..99, rank-exclusive rows elsewhere
    lo = 0 if rank == 0 else 50
    vals = [float(v) for v in range(lo, lo + 100)] * 2
    pdf = pd.DataFrame({"v": vals})
    idf = AnovosFrame.from_pandas(pdf, device="cpu")
    odf, stats = duplicate_detection(ctx, idf, treatment=True)
    kept = sorted(float(x) for x in odf.col("v").data.tolist())
    gathered = dist.all_gather_object(kept)
    if rank == 0:
        allkept = sorted([v for g in gathered for v in g])
        with open(out, "w") as f:
            json.dump({"all": allkept, "unique_rows": float(stats[stats["metric"] == "unique_rows_count"]["value"].iloc[0])}, f)
    td.barrier()
    td.destroy_process_group()


def test_dist_exact_dedup_across_ranks(tmp_path):
    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_dedup_worker, args=(r, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    res = json.load(open(out))
    # global distinct values: 0..149 — every one kept exactly once
    assert res["all"] == [float(v) for v in range(150)]
    assert res["unique_rows"] == 150.0


def _wrappers_worker(rank, world, port, out_path):
    """Exercise every dist wrapper with rank-asymmetric payloads: varlen
    all_gather_tensor (1-D + 2-D + empty-on-one-rank), batched scalar
    reduce, broadcast_."""
    os.environ.update(
        {"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
         "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)}
    )
    import torch.distributed as td

    from anovos_amd.core import dist

    dist.init_from_env(timeout_s=120)
    # 1-D varlen: rank 0 sends 5 values, rank 1 sends 3
    t = torch.arange(5 - 2 * rank, dtype=torch.float64) + 10 * rank
    parts = dist.all_gather_tensor(t)
    gathered_1d = [p.tolist() for p in parts]
    # empty on rank 1
    e = torch.arange(4) if rank == 0 else torch.empty(0, dtype=torch.int64)
    eparts = [p.tolist() for p in dist.all_gather_tensor(e)]
    # 2-D pairs, different row counts
    m = torch.full((2 + rank, 2), float(rank))
    mparts = [p.shape for p in dist.all_gather_tensor(m)]
    # batched scalars
    sc = dist.all_reduce_scalars([float(rank), 1.0, float(rank) * 2], "sum")
    mx = dist.all_reduce_scalars([float(rank)], "max")
    # broadcast tensor
    b = torch.tensor([3.25, -1.5]) if rank == 0 else torch.zeros(2)
    dist.broadcast_(b)
    if rank == 0:
        json.dump(
            {"g1": gathered_1d, "ge": eparts, "gm": [list(s) for s in mparts],
             "sc": sc, "mx": mx, "b": b.tolist(), "backend": dist.backend()},
            open(out_path, "w"),
        )
    td.barrier()
    td.destroy_process_group()


def test_dist_wrapper_primitives_2rank():
    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_wrappers_worker, args=(r, 2, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0
    res = json.load(open(out))
    assert res["backend"] == "gloo"
    assert res["g1"] == [[0.0, 1.0, 2.0, 3.0, 4.0], [10.0, 11.0, 12.0]]
    assert res["ge"] == [[0, 1, 2, 3], []]
    assert res["gm"] == [[2, 2], [3, 2]]
    assert res["sc"] == [1.0, 2.0, 2.0]
    assert res["mx"] == [1.0]
    assert res["b"] == [3.25, -1.5]


def _stable_worker(rank, world, port, out_path):
    os.environ.update(
        {"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": str(world),
         "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)}
    )
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops import histogram as hist_ops
    from anovos_amd.ops import stats as stats_ops
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    init_context("cpu")
    rng = np.random.default_rng(123)
    n = 300_000  # above the exact-sort threshold -> sketch / dense-int paths
    off_full = rng.normal(0, 1, n) + 1e9
    k_full = rng.integers(0, 43, n).astype(np.float32)
    half = n // 2
    sl = slice(rank * half, (rank + 1) * half)
    idf = AnovosFrame(
        {"off": Column("off", "double", torch.tensor(off_full[sl])),
         "k": Column("k", "float", torch.tensor(k_full[sl]))},
        device="cpu",
    )
    m = stats_ops.frame_moments(idf, ["off", "k"])["off"]
    q = hist_ops.approx_quantiles(idf, ["k", "off"], [0.25, 0.5, 0.9])
    if rank == 0:
        with open(out_path, "w") as f:
            json.dump({"stddev": m.stddev, "skew": m.skewness, "kurt": m.kurtosis,
                       "mean": m.mean, "k_q": q["k"], "off_q": q["off"]}, f)
    td.barrier()
    td.destroy_process_group()


def test_dist_stable_moments_and_integral_quantiles():
    """2-rank gloo: the shared accumulation pivot keeps skew/kurt correct
    at |mean| >> stddev, and integral-column quantiles stay EXACT across
    the cross-rank dense-histogram merge."""
    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_stable_worker, args=(r, 2, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=240)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    with open(out) as f:
        res = json.load(f)
    rng = np.random.default_rng(123)
    n = 300_000
    off_full = rng.normal(0, 1, n) + 1e9
    k_full = rng.integers(0, 43, n).astype(np.float32)
    assert res["mean"] == pytest.approx(float(off_full.mean()), rel=1e-12)
    assert res["stddev"] == pytest.approx(float(off_full.std(ddof=1)), rel=1e-3)
    assert abs(res["skew"]) < 0.05 and abs(res["kurt"]) < 0.1
    import math

    ks = np.sort(k_full)
    exact_k = [float(ks[min(max(math.ceil(p * n), 1), n) - 1]) for p in (0.25, 0.5, 0.9)]
    assert res["k_q"] == exact_k  # dense integral path is exact
    for got, p in zip(res["off_q"], (0.25, 0.5, 0.9)):
        assert got == pytest.approx(float(np.quantile(off_full, p)), abs=0.05)


def _mode_worker(rank, port, out):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops import groupby
    from anovos_amd.shared.context import init_context

    dist.init_from_env(timeout_s=120)
    init_context("cpu")
    rng = np.random.default_rng(77)
    n = 2_000_000  # above the partitioned-exchange threshold
    full = rng.normal(0, 1, n)
    # plant the true mode: one value repeated across BOTH shards
    planted = 0.123456789
    full[rng.choice(n, 500, replace=False)] = planted
    half = n // 2
    shard = full[rank * half : (rank + 1) * half]
    idf = AnovosFrame({"x": Column("x", "double", torch.tensor(shard))}, device="cpu")
    res = groupby.discrete_modes(idf, ["x"])["x"]
    if rank == 0:
        json.dump({"mode": res[0], "count": res[1]}, open(out, "w"))
    td.barrier()
    td.destroy_process_group()


def test_dist_exact_continuous_mode_partitioned():
    """High-cardinality continuous column: the exact hash-partitioned
    mode must find the planted duplicate exactly (VERDICT r01 item 7 —
    no histogram approximation)."""
    port = _free_port()
    out = tempfile.NamedTemporaryFile(suffix=".json", delete=False).name
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_mode_worker, args=(r, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0
    res = json.load(open(out))
    assert res["mode"] == pytest.approx(0.123456789, abs=1e-12)
    assert res["count"] == 500


def _workflow_worker(rank, port, workdir, out):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    os.chdir(workdir)
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd import workflow

    dist.init_from_env(timeout_s=180)
    from anovos_amd.shared.context import init_context

    init_context("cpu")
    df = workflow.run(os.path.join(workdir, "cfg.yaml"))
    rows = df.count()  # collective: must run on BOTH ranks
    if rank == 0:
        import pandas as _pd

        counts = _pd.read_csv("report_stats/measures_of_counts.csv")
        json.dump({"rows": rows,
                   "fill_age": float(counts[counts["attribute"] == "age"]["fill_count"].iloc[0]),
                   "report": os.path.exists("report_stats/ml_anovos_report.html")}, open(out, "w"))
    td.barrier()
    td.destroy_process_group()


def test_dist_full_workflow_two_ranks(tmp_path):
    """The entire YAML workflow (ETL -> stats -> QC -> associations ->
    transformers -> report) as TWO ranks over gloo: part files shard by
    rank, stats merge globally, rank 0 writes the report. Mirrors how
    the 8-GPU run executes."""
    import sys as _sys

    import pandas as pd
    import yaml as _yaml

    _sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))
    import make_income_data as mid

    pdf = mid.make(4000)
    d = tmp_path / "data" / "income_dataset" / "csv"
    d.mkdir(parents=True)
    pdf.iloc[:2000].to_csv(d / "part-00000.csv", index=False)
    pdf.iloc[2000:].to_csv(d / "part-00001.csv", index=False)
    cfg = {
        "input_dataset": {
            "read_dataset": {"file_path": "data/income_dataset/csv", "file_type": "csv",
                             "file_configs": {"header": True, "inferSchema": True}},
            "delete_column": ["logfnl"],
        },
        "stats_generator": {
            "metric": ["global_summary", "measures_of_counts", "measures_of_dispersion"],
            "metric_args": {"list_of_cols": "all", "drop_cols": ["ifa"]},
        },
        "quality_checker": {
            "nullColumns_detection": {"list_of_cols": "all", "drop_cols": ["ifa", "income"],
                                       "treatment": True, "treatment_method": "MMM"},
        },
        "association_evaluator": {
            "IV_calculation": {"list_of_cols": "all", "drop_cols": ["ifa"],
                               "label_col": "income", "event_label": ">50K"},
        },
        "transformers": {
            "numerical_rescaling": {"z_standardization": {"list_of_cols": ["age"], "output_mode": "append"}},
        },
        "report_preprocessing": {
            "master_path": "report_stats",
            "charts_to_objects": {"list_of_cols": "all", "drop_cols": ["ifa"],
                                   "label_col": "income", "event_label": ">50K",
                                   "bin_method": "equal_frequency", "bin_size": 10,
                                   "source_path": "inter"},
        },
        "report_generation": {"master_path": "report_stats", "final_report_path": "report_stats",
                              "label_col": "income", "event_label": ">50K"},
    }
    with open(tmp_path / "cfg.yaml", "w") as f:
        _yaml.safe_dump(cfg, f, sort_keys=False)
    port = _free_port()
    out = str(tmp_path / "res.json")
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_workflow_worker, args=(r, port, str(tmp_path), out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"workflow worker failed: exit {p.exitcode}"
    res = json.load(open(out))
    assert res["rows"] == 4000  # both shards merged in the global count
    truth_fill = int(pdf["age"].notna().sum())
    assert res["fill_age"] == truth_fill
    assert res["report"] is True


def test_dist_workflow_fewer_parts_than_ranks(tmp_path):
    """A single-part dataset at world_size=2: rank 1 holds an EMPTY
    shard and every stage (stats, quantiles, checkers, associations,
    report) must still complete with globally correct results — this is
    how an 8-GPU node behaves on small side tables."""
    import sys as _sys

    import pandas as pd
    import yaml as _yaml

    _sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))
    import make_income_data as mid

    pdf = mid.make(3000)
    d = tmp_path / "data" / "income_dataset" / "csv"
    d.mkdir(parents=True)
    pdf.to_csv(d / "part-00000.csv", index=False)  # ONE part only
    cfg = {
        "input_dataset": {
            "read_dataset": {"file_path": "data/income_dataset/csv", "file_type": "csv",
                             "file_configs": {"header": True, "inferSchema": True}},
            "delete_column": ["logfnl"],
        },
        "stats_generator": {
            "metric": ["global_summary", "measures_of_counts", "measures_of_percentiles"],
            "metric_args": {"list_of_cols": "all", "drop_cols": ["ifa"]},
        },
        "quality_checker": {
            "outlier_detection": {"list_of_cols": "all", "drop_cols": ["ifa", "income"],
                                   "detection_side": "both", "treatment": False},
        },
        # intermediate write + sharded re-read: rank 1 writes a header-only
        # part and re-infers every column as string — the schema must be
        # reconciled across ranks or the dictionary-unify collective
        # count diverges (r02 deadlock)
        "write_intermediate": {
            "file_path": "intermediate_data",
            "file_type": "csv",
            "file_configs": {"mode": "overwrite", "header": True, "delimiter": ",",
                              "inferSchema": True},
        },
        "association_evaluator": {
            "IV_calculation": {"list_of_cols": "all", "drop_cols": ["ifa"],
                               "label_col": "income", "event_label": ">50K"},
        },
        "report_preprocessing": {
            "master_path": "report_stats",
            "charts_to_objects": {"list_of_cols": "all", "drop_cols": ["ifa"],
                                   "label_col": "income", "event_label": ">50K",
                                   "bin_method": "equal_frequency", "bin_size": 10,
                                   "source_path": "inter"},
        },
        "report_generation": {"master_path": "report_stats", "final_report_path": "report_stats",
                              "label_col": "income", "event_label": ">50K"},
    }
    with open(tmp_path / "cfg.yaml", "w") as f:
        _yaml.safe_dump(cfg, f, sort_keys=False)
    port = _free_port()
    out = str(tmp_path / "res.json")
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_workflow_worker, args=(r, port, str(tmp_path), out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    res = json.load(open(out))
    assert res["rows"] == 3000
    assert res["fill_age"] == int(pdf["age"].notna().sum())
    assert res["report"] is True


def _cfg_worker(rank, port, workdir, out, artifacts):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    os.chdir(workdir)
    import torch.distributed as td

    from anovos_amd import workflow
    from anovos_amd.core import dist

    dist.init_from_env(timeout_s=180)
    from anovos_amd.shared.context import init_context

    init_context("cpu")
    df = workflow.run(os.path.join(workdir, "cfg.yaml"))
    rows = df.count()  # collective: must run on BOTH ranks
    if rank == 0:
        json.dump({"rows": rows, "artifacts": {a: os.path.exists(a) for a in artifacts}},
                  open(out, "w"))
    td.barrier()
    td.destroy_process_group()


def _run_cfg_two_ranks(tmp_path, cfg, artifacts):
    import yaml as _yaml

    with open(tmp_path / "cfg.yaml", "w") as f:
        _yaml.safe_dump(cfg, f, sort_keys=False)
    port = _free_port()
    out = str(tmp_path / "res.json")
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_cfg_worker, args=(r, port, str(tmp_path), out, artifacts))
             for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=300)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    return json.load(open(out))


def test_dist_workflow_timeseries_two_ranks(tmp_path):
    """The ts-config workflow at world_size=2 with rank-DISJOINT date
    ranges per shard (sorted before splitting): timestamp auto-detection
    decisions and the viz top-category choices must be reconciled
    globally or the per-column collective loops deadlock (r02 bug)."""
    import sys as _sys

    _sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))
    import make_income_data as mid

    pdf = mid.add_ts_cols(mid.make(3000))
    pdf = pdf.sort_values("txn_date").reset_index(drop=True)  # disjoint dates per part
    d = tmp_path / "data" / "income_dataset" / "csv"
    d.mkdir(parents=True)
    pdf.iloc[:1500].to_csv(d / "part-00000.csv", index=False)
    pdf.iloc[1500:].to_csv(d / "part-00001.csv", index=False)
    cfg = {
        "input_dataset": {
            "read_dataset": {"file_path": "data/income_dataset/csv", "file_type": "csv",
                             "file_configs": {"header": True, "inferSchema": True}},
            "delete_column": ["logfnl"],
        },
        "timeseries_analyzer": {"auto_detection": True, "id_col": "ifa", "tz_offset": "local",
                                 "inspection": True, "analysis_level": "daily", "max_days": 3600},
        "stats_generator": {"metric": ["global_summary", "measures_of_counts"],
                            "metric_args": {"list_of_cols": "all", "drop_cols": ["ifa"]}},
        "report_preprocessing": {"master_path": "report_stats"},
    }
    res = _run_cfg_two_ranks(tmp_path, cfg, [
        "report_stats/ts_cols_stats.csv",
        "report_stats/stats_txn_date_1.csv",
        "report_stats/txn_date_age_daily.csv",
    ])
    assert res["rows"] == 3000
    assert all(res["artifacts"].values()), res["artifacts"]
    import pandas as _pd

    ts_stats = _pd.read_csv(tmp_path / "report_stats" / "ts_cols_stats.csv")
    assert "txn_date" in set(ts_stats["attribute"])


def test_dist_workflow_geospatial_two_ranks(tmp_path):
    """The geo-config workflow at world_size=2: lat/long/geohash
    detection reconciled across ranks, clustering on the gathered
    global sample (rank 1's shard alone once crashed k-means), and the
    per-row-unique geohash dictionary healed by align_dictionaries
    before dictionary-indexed collectives (r02 bugs)."""
    import sys as _sys

    _sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "tools"))
    import make_income_data as mid

    pdf = mid.add_geo_cols(mid.make(2400))
    d = tmp_path / "data" / "income_dataset" / "csv"
    d.mkdir(parents=True)
    pdf.iloc[:1200].to_csv(d / "part-00000.csv", index=False)
    pdf.iloc[1200:].to_csv(d / "part-00001.csv", index=False)
    cfg = {
        "input_dataset": {
            "read_dataset": {"file_path": "data/income_dataset/csv", "file_type": "csv",
                             "file_configs": {"header": True, "inferSchema": True}},
            "delete_column": ["logfnl"],
        },
        "geospatial_controller": {
            "geospatial_analyzer": {"auto_detection_analyzer": True, "id_col": "ifa",
                                     "max_analysis_records": 10000, "top_geo_records": 50,
                                     "max_cluster": 6, "eps": "0.3,0.5,0.1",
                                     "min_samples": "40,120,40", "global_map_box_val": 0},
        },
        "stats_generator": {"metric": ["global_summary", "measures_of_counts"],
                            "metric_args": {"list_of_cols": "all", "drop_cols": ["ifa"]}},
        "report_preprocessing": {"master_path": "report_stats"},
    }
    res = _run_cfg_two_ranks(tmp_path, cfg, [
        "report_stats/cluster_output_kmeans_latitude_longitude.csv",
        "report_stats/Overall_Summary_2_gh7.csv",
        "report_stats/measures_of_counts.csv",
    ])
    assert res["rows"] == 2400
    assert all(res["artifacts"].values()), res["artifacts"]


def _align_dict_worker(rank, port, out):
    os.environ.update({"RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
                       "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port)})
    import torch.distributed as td

    from anovos_amd.core import dist
    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops.groupby import align_dictionaries, cat_value_counts

    dist.init_from_env(timeout_s=120)
    # columns built AFTER ingest (transformer outputs) carry per-rank
    # dictionaries; rank 0: a,b,c / rank 1: b,d (+ a null)
    if rank == 0:
        codes = torch.tensor([0, 0, 1, 2, 1], dtype=torch.int32)
        d = ["a", "b", "c"]
    else:
        codes = torch.tensor([0, 1, 1, -1], dtype=torch.int32)
        d = ["b", "d"]
    idf = AnovosFrame({"g": Column("g", "string", codes, d)}, device="cpu")
    counts = cat_value_counts(idf, ["g"])["g"]
    col = idf.col("g")
    res = {v: int(c) for v, c in zip(col.dictionary, counts.tolist())}
    align_dictionaries(idf, ["g"])  # idempotent second call
    same = list(col.dictionary)
    if rank == 0:
        json.dump({"counts": res, "dict": same}, open(out, "w"))
    td.barrier()
    td.destroy_process_group()


def test_dist_align_dictionaries(tmp_path):
    """Post-ingest categorical columns with rank-divergent dictionaries
    are healed in place before the dictionary-indexed all-reduce
    (regression: per-row-unique geohash dictionaries sheared the
    collective)."""
    port = _free_port()
    out = str(tmp_path / "res.json")
    mp_ctx = mp.get_context("spawn")
    procs = [mp_ctx.Process(target=_align_dict_worker, args=(r, port, out)) for r in range(2)]
    for p in procs:
        p.start()
    for p in procs:
        p.join(timeout=180)
        assert p.exitcode == 0, f"worker failed: exit {p.exitcode}"
    res = json.load(open(out))
    assert res["dict"] == ["a", "b", "c", "d"]
    # global truth: a=2, b=2+2=... rank0 b codes: idx1 x2 -> 2; rank1 b: idx0 x1 -> 1
    assert res["counts"] == {"a": 2, "b": 3, "c": 1, "d": 2}
