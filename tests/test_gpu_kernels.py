"""GPU numerics tests: every HIP kernel vs a plain PyTorch fp32/fp64
reference of the same op (run with `pytest -m gpu` on an MI355X)."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@pytest.fixture(scope="module")
def ext():
    from anovos_amd.ops import backend

    e = backend.require_hip()
    assert e is not None, "HIP extension must be loadable on the GPU box"
    return e


@pytest.fixture(scope="module")
def gpu_cols():
    g = torch.Generator().manual_seed(3)
    cols = []
    for i in range(8):
        x = torch.randn(1_000_000 + i * 1000, generator=g) * (i + 1) + i
        x[torch.rand(x.shape[0], generator=g) < 0.02] = float("nan")
        cols.append(x.cuda())
    cols.append((torch.randn(500_000, generator=g).double() * 3).cuda())  # fp64 col
    return cols


@requires_gpu
def test_column_moments_vs_torch(ext, gpu_cols):
    out = ext.column_moments(gpu_cols).cpu()
    for i, t in enumerate(gpu_cols):
        td = t.cpu().to(torch.float64)
        valid = ~torch.isnan(td)
        x = td[valid]
        n, s1, s2, s3, s4, mn, mx, zn, nf = out[i].tolist()
        assert n == int(valid.sum())
        assert abs(s1 - float(x.sum())) <= 1e-9 * max(abs(float(x.sum())), 1)
        assert abs(s2 - float((x * x).sum())) <= 1e-9 * float((x * x).sum())
        assert abs(s4 - float((x**4).sum())) <= 1e-8 * float((x**4).sum())
        assert mn == pytest.approx(float(x.min()), rel=0, abs=0)
        assert mx == pytest.approx(float(x.max()), rel=0, abs=0)


@requires_gpu
def test_column_histograms_vs_torch(ext, gpu_cols):
    cols = gpu_cols[:4]
    lo = torch.tensor([float(torch.nanquantile(t[:100000].float().cpu(), 0.0)) for t in cols], dtype=torch.float64)
    lo = torch.tensor([float(t[~torch.isnan(t)].min()) for t in cols], dtype=torch.float64)
    hi = torch.tensor([float(t[~torch.isnan(t)].max()) for t in cols], dtype=torch.float64)
    nbins = 512
    out = ext.column_histograms(cols, lo.cuda(), hi.cuda(), nbins).cpu()
    for i, t in enumerate(cols):
        x = t.cpu().to(torch.float64)
        x = x[~torch.isnan(x)]
        idx = ((x - lo[i]) * (nbins / (hi[i] - lo[i]))).long().clamp_(0, nbins - 1)
        ref = torch.bincount(idx, minlength=nbins)
        assert int(out[i].sum()) == x.numel()
        assert torch.equal(out[i], ref)


@requires_gpu
def test_bucketize_vs_torch(ext, gpu_cols):
    cols = gpu_cols[:3]
    cuts = [torch.sort(torch.randn(9, generator=torch.Generator().manual_seed(i)))[0].to(torch.float64).cuda() for i in range(3)]
    outs = ext.bucketize_columns(cols, cuts)
    for t, c, o in zip(cols, cuts, outs):
        ref = torch.bucketize(t.to(torch.float64), c, right=False).to(torch.int32)
        ref = torch.where(torch.isnan(t), torch.full_like(ref, -1), ref)
        assert torch.equal(o.cpu(), ref.cpu())


@requires_gpu
def test_code_counts_vs_torch(ext):
    g = torch.Generator().manual_seed(9)
    codes = torch.randint(-1, 1000, (2_000_000,), generator=g).to(torch.int32).cuda()
    out = ext.code_counts(codes, 1000).cpu()
    valid = codes.cpu()[codes.cpu() >= 0].long()
    ref = torch.bincount(valid, minlength=1000)
    assert torch.equal(out, ref)


@requires_gpu
def test_hll_vs_cpu_reference(ext):
    from anovos_amd.ops import distinct as distinct_ops

    g = torch.Generator().manual_seed(5)
    x = torch.randint(0, 50_000, (3_000_000,), generator=g).to(torch.float32)
    regs_gpu = ext.hll_registers(x.cuda(), distinct_ops.HLL_P).cpu()
    regs_cpu = distinct_ops.hll_registers(x)  # torch reference path
    assert torch.equal(regs_gpu.to(torch.int32), regs_cpu.to(torch.int32))
    est = distinct_ops.hll_estimate(regs_gpu)
    exact = int(torch.unique(x).numel())
    assert abs(est - exact) / exact < 0.03


@requires_gpu
def test_row_null_counts_vs_torch(ext, gpu_cols):
    n = min(t.numel() for t in gpu_cols[:5])
    cols = [t[:n].contiguous() for t in gpu_cols[:5]]
    out = torch.zeros(n, dtype=torch.int32, device="cuda")
    ext.row_null_counts_num(cols, out)
    ref = torch.zeros(n, dtype=torch.int32)
    for t in cols:
        ref += torch.isnan(t.cpu()).to(torch.int32)
    assert torch.equal(out.cpu(), ref)


@requires_gpu
def test_gpu_pipeline_stats_match_cpu(ext):
    """End-to-end: stats computed on GPU (HIP kernels) match the CPU
    torch reference paths on the same data."""
    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_analyzer import stats_generator as sg
    from anovos_amd.shared.context import AnovosContext
    import pandas as pd

    rng = np.random.default_rng(17)
    n = 200_000
    pdf = pd.DataFrame(
        {
            "a": rng.normal(10, 3, n),
            "b": rng.integers(0, 50, n).astype(float),
            "c": rng.choice(["x", "y", "z"], n),
        }
    )
    pdf.loc[rng.choice(n, 1000, replace=False), "a"] = np.nan
    cpu_f = AnovosFrame.from_pandas(pdf, device="cpu")
    gpu_f = cpu_f.to_device("cuda:0")
    ctx_gpu = AnovosContext("cuda:0")
    ctx_cpu = AnovosContext("cpu")

    for fn in [sg.measures_of_counts, sg.measures_of_dispersion, sg.measures_of_shape]:
        a = fn(ctx_cpu, cpu_f).set_index("attribute")
        b = fn(ctx_gpu, gpu_f).set_index("attribute")
        for col in a.columns:
            for attr in a.index:
                va, vb = a.loc[attr, col], b.loc[attr, col]
                if va is None or (isinstance(va, float) and va != va):
                    continue
                assert abs(float(va) - float(vb)) <= max(1e-3 * abs(float(va)), 1e-3), (fn.__name__, attr, col, va, vb)
    pa = sg.measures_of_percentiles(ctx_cpu, cpu_f).set_index("attribute")
    pb = sg.measures_of_percentiles(ctx_gpu, gpu_f).set_index("attribute")
    for attr in pa.index:
        for col in ["25%", "50%", "75%", "95%"]:
            assert abs(float(pa.loc[attr, col]) - float(pb.loc[attr, col])) <= max(
                0.01 * abs(float(pa.loc[attr, col])), 0.05
            )


@requires_gpu
def test_hll_multi_matches_single(ext, gpu_cols):
    cols = [t for t in gpu_cols if t.dtype == torch.float32][:4]
    multi = ext.hll_registers_multi(cols, 14).cpu()
    for i, t in enumerate(cols):
        single = ext.hll_registers(t, 14).cpu()
        assert torch.equal(multi[i], single)


@requires_gpu
def test_scale_columns_vs_torch(ext, gpu_cols):
    cols = [t for t in gpu_cols if t.dtype == torch.float32][:4]
    a = [1.0, -2.0, 0.5, 3.0]
    b = [2.0, 0.25, -1.0, 10.0]
    outs = ext.scale_columns(cols, torch.tensor(a), torch.tensor(b))
    for t, ai, bi, o in zip(cols, a, b, outs):
        ref = (t - ai) * bi
        assert torch.allclose(o, ref, rtol=1e-6, atol=1e-5, equal_nan=True)


@requires_gpu
def test_fill_nan_columns_vs_torch(ext, gpu_cols):
    cols = [t for t in gpu_cols if t.dtype == torch.float32][:4]
    fills = [0.0, 1.5, -3.0, 42.0]
    outs = ext.fill_nan_columns(cols, torch.tensor(fills))
    for t, f, o in zip(cols, fills, outs):
        ref = torch.nan_to_num(t, nan=f)
        assert torch.equal(o, ref)


@requires_gpu
def test_centered_gram_bf16_vs_torch(ext):
    """K8 MFMA Gram vs fp32 torch reference (bf16 input tolerance)."""
    g = torch.Generator(device="cpu").manual_seed(5)
    n, k = 200_000, 37  # k not a multiple of 16 — exercises tile padding
    cols = []
    for i in range(k):
        x = torch.randn(n, generator=g) * (1 + i % 5) + i
        x[torch.rand(n, generator=g) < 0.01] = float("nan")
        cols.append(x.cuda())
    means = torch.tensor([torch.nanmean(c).item() for c in cols], device="cuda")
    gram = ext.centered_gram_bf16([c.contiguous() for c in cols], means)
    X = torch.stack([torch.nan_to_num(c, nan=float(means[i])) for i, c in enumerate(cols)], dim=1)
    Xc = X - means.unsqueeze(0)
    ref = (Xc.T @ Xc).float()
    scale = ref.abs().max()
    assert torch.allclose(gram, ref, atol=float(scale) * 2e-2), float((gram - ref).abs().max() / scale)
    # symmetry is exact
    assert torch.equal(gram, gram.T)


@requires_gpu
def test_pearson_matrix_gpu_path(ext):
    """ops.corr.pearson_matrix via the MFMA kernel vs numpy corrcoef."""
    import numpy as np

    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops.corr import pearson_matrix

    g = torch.Generator(device="cpu").manual_seed(6)
    n = 100_000
    base = torch.randn(n, generator=g)
    cols = {
        "a": Column("a", "float", base.cuda()),
        "b": Column("b", "float", (base * 2 + torch.randn(n, generator=g) * 0.1).cuda()),
        "c": Column("c", "float", torch.randn(n, generator=g).cuda()),
    }
    idf = AnovosFrame(cols, device="cuda")
    corr = pearson_matrix(idf, ["a", "b", "c"])
    X = np.stack([cols[c].data.cpu().numpy() for c in "abc"])
    ref = np.corrcoef(X)
    assert np.allclose(corr, ref, atol=5e-3), np.abs(corr - ref).max()


@requires_gpu
def test_code_counts_multi_vs_bincount(ext):
    g = torch.Generator(device="cpu").manual_seed(9)
    sizes = [5, 40, 1]
    cols = []
    for s in sizes:
        c = torch.randint(0, s, (100_000,), generator=g, dtype=torch.int32)
        c[torch.rand(100_000, generator=g) < 0.05] = -1  # nulls
        cols.append(c.cuda())
    flat = ext.code_counts_multi(cols, sizes)
    off = 0
    for c, s in zip(cols, sizes):
        cnt = flat[off : off + s].cpu()
        nulls = int(flat[off + s])
        ref = torch.bincount(c[c >= 0].long().cpu(), minlength=s)
        assert torch.equal(cnt, ref)
        assert nulls == int((c < 0).sum())
        off += s + 1


@requires_gpu
def test_outlier_clamp_columns_vs_torch(ext):
    g = torch.Generator(device="cpu").manual_seed(10)
    n = 100_000
    cols = [torch.randn(n, generator=g).cuda() * 10 for _ in range(4)]
    cols[1][::100] = float("nan")
    lo = torch.tensor([-5.0, float("nan"), -3.0, -1.0])
    hi = torch.tensor([5.0, 8.0, float("nan"), 1.0])
    counts, outs = ext.outlier_clamp_columns([c.contiguous() for c in cols], lo, hi, 1)
    for i, c in enumerate(cols):
        l, h = float(lo[i]), float(hi[i])
        m = ~torch.isnan(c)
        exp_lo = int(((c < l) & m).sum()) if l == l else 0
        exp_hi = int(((c > h) & m).sum()) if h == h else 0
        assert int(counts[i, 0]) == exp_lo, i
        assert int(counts[i, 1]) == exp_hi, i
        ref = c.clone()
        if l == l:
            ref = torch.where((ref < l) & m, torch.full_like(ref, l), ref)
        if h == h:
            ref = torch.where((ref > h) & m, torch.full_like(ref, h), ref)
        assert torch.allclose(outs[i], ref, rtol=0, atol=0, equal_nan=True), i
    # mode 2: null replacement
    counts2, outs2 = ext.outlier_clamp_columns([cols[0].contiguous()], lo[:1], hi[:1], 2)
    x = cols[0]
    nulled = torch.isnan(outs2[0]).sum()
    assert int(nulled) == int(((x < -5) | (x > 5)).sum())


@requires_gpu
def test_approx_quantiles_sketch_path_vs_torch(ext):
    """Large-n path: pass-1 histogram + LUT-grouped refinement must land
    within Spark's rel-err 0.01 rank tolerance of torch.quantile."""
    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops import histogram as hist_ops

    g = torch.Generator(device="cpu").manual_seed(23)
    n = 2_000_000  # >> EXACT_N_THRESHOLD -> sketch + refinement path
    cols = {}
    specs = {
        "normal": torch.randn(n, generator=g) * 7 + 3,
        "lognormal": torch.randn(n, generator=g).exp() * 10,
        "uniform": torch.rand(n, generator=g) * 1000 - 500,
        "heavy_dup": torch.randint(0, 20, (n,), generator=g).to(torch.float32),
    }
    for name, x in specs.items():
        x[torch.rand(n, generator=g) < 0.01] = float("nan")
        cols[name] = Column(name, "float", x.cuda())
    idf = AnovosFrame(cols, device="cuda")
    probs = [0.01, 0.05, 0.25, 0.5, 0.75, 0.95, 0.99]
    got = hist_ops.approx_quantiles(idf, list(specs), probs)
    for name in specs:
        x = cols[name].data
        v = x[~torch.isnan(x)]
        vs, _ = torch.sort(v)
        nn = vs.numel()
        for p, q in zip(probs, got[name]):
            # rank-tolerance check (Spark approxQuantile contract): the
            # returned value must sit next to a data value whose rank
            # interval covers p*n within rel_err (duplicate blocks make a
            # single-point rank ambiguous)
            qt = torch.tensor(q, device=vs.device)
            lo_r = int(torch.searchsorted(vs, qt, right=False))
            hi_r = int(torch.searchsorted(vs, qt, right=True))
            # rank intervals of the data values bracketing q
            cand = [(lo_r, hi_r)]
            if lo_r > 0:
                v_below = vs[lo_r - 1]
                cand.append((int(torch.searchsorted(vs, v_below, right=False)),
                             int(torch.searchsorted(vs, v_below, right=True))))
            if hi_r < nn:
                v_above = vs[hi_r]
                cand.append((int(torch.searchsorted(vs, v_above, right=False)),
                             int(torch.searchsorted(vs, v_above, right=True))))
            tol = max(0.01 * nn, 1000)
            target = p * nn
            ok = any(a - tol <= target <= b + tol for a, b in cand)
            assert ok, (name, p, q, cand, target)


@requires_gpu
def test_moments_hll_fused_vs_separate(ext):
    """Fused K1/K2+K4 must match the separate moments and HLL kernels."""
    g = torch.Generator(device="cpu").manual_seed(31)
    cols = []
    for i in range(5):
        x = torch.randn(500_000, generator=g) * (i + 1)
        x[torch.rand(500_000, generator=g) < 0.02] = float("nan")
        cols.append(x.cuda().contiguous())
    mom_f, regs_f = ext.moments_hll(cols, 12)
    mom_s = ext.column_moments(cols)
    regs_s = ext.hll_registers_multi(cols, 12)
    assert torch.allclose(mom_f.cpu(), mom_s.cpu(), rtol=1e-12, atol=1e-9, equal_nan=True)
    assert torch.equal(regs_f.cpu(), regs_s.cpu())


@requires_gpu
def test_row_null_counts_mixed_dtypes(ext):
    """Fused row-null covers float32 NaN, float64 NaN AND int32 -1 codes."""
    g = torch.Generator(device="cpu").manual_seed(41)
    n = 300_000
    f32 = torch.randn(n, generator=g)
    f32[torch.rand(n, generator=g) < 0.1] = float("nan")
    f64 = torch.randn(n, generator=g).double()
    f64[torch.rand(n, generator=g) < 0.05] = float("nan")
    codes = torch.randint(-1, 10, (n,), generator=g).to(torch.int32)
    cols = [f32.cuda().contiguous(), f64.cuda().contiguous(), codes.cuda().contiguous()]
    out = torch.zeros(n, dtype=torch.int32, device="cuda")
    ext.row_null_counts_num(cols, out)
    ref = torch.isnan(f32).to(torch.int32) + torch.isnan(f64).to(torch.int32) + (codes == -1).to(torch.int32)
    assert torch.equal(out.cpu(), ref)


@requires_gpu
def test_captured_transform_graph_replay(ext):
    """hipGraph capture of a transform chain replays correctly on new
    batches (fixed shapes)."""
    import time

    from anovos_amd.ops.graph import CapturedTransform

    g = torch.Generator(device="cpu").manual_seed(55)
    n, k = 1_000_000, 8

    def chain(cols):
        # scaler + clamp + fill: a typical fitted transform apply
        out = []
        for i, t in enumerate(cols):
            y = (t - float(i)) * 0.5
            y = torch.clamp(y, -3.0, 3.0)
            y = torch.nan_to_num(y, nan=0.0)
            out.append(y)
        return out

    example = [torch.randn(n, generator=g).cuda() for _ in range(k)]
    cap = CapturedTransform(chain, example)
    batch = [torch.randn(n, generator=g).cuda() for _ in range(k)]
    batch[0][::10] = float("nan")
    got = cap(batch)
    ref = chain(batch)
    for a, b in zip(got, ref):
        assert torch.allclose(a, b, equal_nan=True)
    # replay must be at least as fast as eager dispatch
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(20):
        cap(batch)
    torch.cuda.synchronize()
    graph_t = time.perf_counter() - t0
    t0 = time.perf_counter()
    for _ in range(20):
        chain(batch)
    torch.cuda.synchronize()
    eager_t = time.perf_counter() - t0
    assert graph_t < eager_t * 1.5  # copies included; replay must not regress


@requires_gpu
def test_column_moments_shifted_stability(ext):
    """Shifted accumulation keeps skew/kurt correct at |mean| >> stddev
    (raw power sums lose every digit at offset 1e8 — SURVEY §6 hard part)."""
    from anovos_amd.ops.stats import MomentStats, compute_column_shifts

    g = torch.Generator(device="cpu").manual_seed(41)
    for offset in (1e6, 1e8, 1e10):
        x = (torch.randn(1_000_000, generator=g, dtype=torch.float64) + offset).cuda()
        shifts = compute_column_shifts([x])
        assert abs(shifts[0] - offset) < 100.0  # pivot lands inside the data
        vec = ext.column_moments([x], shifts).cpu()[0].tolist()
        m = MomentStats(vec, shift=shifts[0])
        assert m.mean == pytest.approx(offset, rel=1e-9)
        assert m.stddev == pytest.approx(1.0, rel=1e-2)
        assert abs(m.skewness) < 0.05
        assert abs(m.kurtosis) < 0.1
        # fused moments+HLL kernel honors the same pivot
        mom_f, _ = ext.moments_hll([x], 12, shifts)
        assert torch.allclose(mom_f.cpu()[0], torch.tensor(vec, dtype=torch.float64), rtol=1e-12, atol=1e-9)


@requires_gpu
def test_pearson_mfma_large_offset():
    """f64 columns with |mean| >> spread keep their correlation through
    the bf16 MFMA path (centered in f64 before the downcast)."""
    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops import corr

    g = torch.Generator(device="cpu").manual_seed(43)
    n = 1_000_000
    z = torch.randn(n, generator=g, dtype=torch.float64)
    a = (z + 1e9).cuda()
    b = (0.5 * z + 0.866 * torch.randn(n, generator=g, dtype=torch.float64) + 1e9).cuda()
    idf = AnovosFrame({"a": Column("a", "double", a), "b": Column("b", "double", b)}, device="cuda")
    m = corr.pearson_matrix(idf, ["a", "b"])
    assert abs(m[0, 1] - 0.5) < 0.02


@requires_gpu
def test_fill_code_columns(ext):
    """Fused categorical null-fill matches the where() reference."""
    g = torch.Generator(device="cpu").manual_seed(47)
    cols, fills = [], []
    for i in range(4):
        n = 1_000_000 + i * 3 + 1  # odd tails exercise the scalar loop
        c = torch.randint(0, 30, (n,), generator=g, dtype=torch.int32)
        c[torch.rand(n, generator=g) < 0.05] = -1
        cols.append(c.cuda().contiguous())
        fills.append(i + 1)
    outs = ext.fill_code_columns(cols, fills)
    for c, f, o in zip(cols, fills, outs):
        ref = torch.where(c == -1, torch.full_like(c, f), c)
        assert torch.equal(o, ref)
        assert int((o == -1).sum()) == 0


@requires_gpu
def test_label_counts_multi_vs_torch(ext):
    """Fused K9 label-conditioned histograms match the torch scatter_add
    reference for mixed f32-binned and int32-code columns."""
    g = torch.Generator(device="cpu").manual_seed(53)
    n = 1_000_003  # odd tail exercises the scalar loop
    label = (torch.rand(n, generator=g) < 0.3).to(torch.uint8)
    # f32 binned column: bins 1..10 + NaN nulls
    b = torch.randint(1, 11, (n,), generator=g).to(torch.float32)
    b[torch.rand(n, generator=g) < 0.03] = float("nan")
    # int32 dictionary codes 0..39 + -1 nulls
    c = torch.randint(0, 40, (n,), generator=g, dtype=torch.int32)
    c[torch.rand(n, generator=g) < 0.05] = -1
    sizes = [12, 41]
    flat = ext.label_counts_multi([b.cuda().contiguous(), c.cuda().contiguous()],
                                  label.cuda().contiguous(), sizes)
    flat = flat.cpu().numpy()
    lab = label.to(torch.float64)
    # reference: python slot mapping (association_evaluator fallback)
    codes_b = (torch.nan_to_num(b, nan=-1.0).to(torch.long) + 1).clamp(min=0)
    codes_c = torch.where(c.to(torch.long) == -1, torch.full((n,), 40, dtype=torch.long), c.to(torch.long))
    off = 0
    for codes, s in ((codes_b, 12), (codes_c, 41)):
        tot_ref = torch.zeros(s, dtype=torch.float64).scatter_add_(0, codes, torch.ones(n, dtype=torch.float64))
        evt_ref = torch.zeros(s, dtype=torch.float64).scatter_add_(0, codes, lab)
        assert np.array_equal(flat[off : off + s], tot_ref.numpy())
        assert np.array_equal(flat[off + s : off + 2 * s], evt_ref.numpy())
        off += 2 * s


@requires_gpu
def test_iv_ig_gpu_matches_cpu(ext):
    """IV/IG through the fused K9 kernel equal the CPU fallback path."""
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_analyzer import association_evaluator as ae
    from anovos_amd.shared.context import init_context

    rng = np.random.default_rng(11)
    n = 200_000
    pdf = pd.DataFrame(
        {
            "x": rng.normal(0, 1, n),
            "y": rng.lognormal(1, 0.4, n),
            "cat": rng.choice(["a", "b", "c", "d"], n),
            "label": rng.choice(["0", "1"], n, p=[0.7, 0.3]),
        }
    )
    pdf.loc[rng.choice(n, 500, replace=False), "x"] = np.nan
    ctx_c = init_context("cpu")
    iv_c = ae.IV_calculation(ctx_c, AnovosFrame.from_pandas(pdf, device="cpu"), label_col="label", event_label="1")
    ig_c = ae.IG_calculation(ctx_c, AnovosFrame.from_pandas(pdf, device="cpu"), label_col="label", event_label="1")
    ctx_g = init_context("cuda")  # leave the global ctx on the GPU for later tests
    iv_g = ae.IV_calculation(ctx_g, AnovosFrame.from_pandas(pdf, device="cuda"), label_col="label", event_label="1")
    ig_g = ae.IG_calculation(ctx_g, AnovosFrame.from_pandas(pdf, device="cuda"), label_col="label", event_label="1")
    for a, b in ((iv_g, iv_c), (ig_g, ig_c)):
        m = a.merge(b, on="attribute", suffixes=("_g", "_c"))
        col = "iv" if "iv_g" in m.columns else "ig"
        assert np.allclose(m[col + "_g"], m[col + "_c"], rtol=1e-4, atol=1e-8), m


@requires_gpu
def test_refinement_many_brackets_and_collisions(ext):
    """>16 refinement brackets per column and brackets
    sharing a pass-1 bin must degrade to extra launches, not crash or
    silently zero. A spiky distribution + 39 quantile probs forces both;
    results must match the exact sort."""
    import math

    from anovos_amd.core.frame import AnovosFrame, Column
    from anovos_amd.ops import histogram as hist_ops

    g = torch.Generator(device="cpu").manual_seed(61)
    n = 2_000_000  # above the exact-sort threshold -> sketch path
    # spiky: 95% of mass on 5 discrete values, the rest uniform
    spikes = torch.tensor([1.0, 2.0, 3.0, 5.0, 8.0])[torch.randint(0, 5, (n,), generator=g)]
    u = torch.rand(n, generator=g) * 10
    x = torch.where(torch.rand(n, generator=g) < 0.95, spikes, u).cuda()
    idf = AnovosFrame({"x": Column("x", "float", x)}, device="cuda")
    probs = [round(0.025 * j, 4) for j in range(1, 40)]  # 39 probs
    got = hist_ops.approx_quantiles(idf, ["x"], probs)["x"]
    xs, _ = torch.sort(x)
    for p, v in zip(probs, got):
        r = min(max(math.ceil(p * n), 1), n) - 1
        exact = float(xs[r])
        assert abs(v - exact) <= max(0.01 * abs(exact), 0.01), (p, v, exact)


@requires_gpu
def test_advanced_imputers_on_device(ext):
    """K15/K13 applies run device-resident and agree with the CPU path."""
    import pandas as pd

    from anovos_amd.core.frame import AnovosFrame
    from anovos_amd.data_transformer import transformers_advanced as TA
    from anovos_amd.shared.context import init_context

    rng = np.random.default_rng(23)
    n = 500_000
    pdf = pd.DataFrame(
        {
            "a": rng.normal(10, 3, n),
            "b": rng.normal(-5, 2, n),
            "c": rng.normal(0, 1, n),
        }
    )
    pdf["b"] = pdf["a"] * 0.5 + rng.normal(0, 0.3, n)
    for c in pdf.columns:
        pdf.loc[rng.choice(n, n // 100, replace=False), c] = np.nan
    import tempfile

    ctx = init_context("cuda")
    gpu_f = AnovosFrame.from_pandas(pdf, device="cuda")
    cpu_f = AnovosFrame.from_pandas(pdf, device="cpu")
    for method in ("KNN", "regression"):
        # fit ONCE (CPU), share the pickled model: isolates the APPLY
        # path (device sampling differs, so independent fits would
        # legitimately produce different models)
        mp = tempfile.mkdtemp(prefix=f"imp_{method}_")
        TA.imputation_sklearn(ctx, cpu_f, "all", method_type=method, sample_size=4000, model_path=mp)
        g = TA.imputation_sklearn(ctx, gpu_f, "all", method_type=method, pre_existing_model=True, model_path=mp)
        c0 = TA.imputation_sklearn(ctx, cpu_f, "all", method_type=method, pre_existing_model=True, model_path=mp)
        for col in pdf.columns:
            gv = g.col(col).data.cpu().numpy()
            cv = c0.col(col).data.numpy()
            assert not np.isnan(gv).any()
            # GEMM reduction order differs between devices; near-tie
            # neighbor choices may flip on a handful of rows
            mism = ~np.isclose(gv, cv, rtol=1e-4, atol=1e-4)
            assert mism.mean() < 1e-3, (method, col, float(mism.mean()))
    mf = TA.imputation_matrixFactorization(ctx, gpu_f, "all", rank=3, max_iter=8)
    for col in pdf.columns:
        assert not torch.isnan(mf.col(col).data).any()


@requires_gpu
def test_bucketize_label_counts_edges(ext):
    """Fused K6+K9 edge conditions: empty cutoffs, f64 input, all-null
    column, all-events label, slots exactly 16 (private-counter bound)."""
    n = 1_000_001
    g = torch.Generator(device="cpu").manual_seed(71)
    lab1 = torch.ones(n, dtype=torch.uint8).cuda()
    x = torch.randn(n, generator=g).cuda()
    # empty cutoffs: every non-null value lands in bin 1 -> slot 2
    flat = ext.bucketize_label_counts([x.contiguous()], [torch.empty(0, dtype=torch.float64)], lab1, [4]).cpu()
    assert int(flat[2]) == n and int(flat[4 + 2]) == n
    # all-null f32 column -> slot 0
    xn = torch.full((n,), float("nan")).cuda()
    flat = ext.bucketize_label_counts([xn], [torch.tensor([0.0])], lab1, [4]).cpu()
    assert int(flat[0]) == n and int(flat[4 + 0]) == n
    # f64 column vs torch reference, 14 cutoffs -> slots 16 exactly
    xd = (torch.randn(n, generator=g, dtype=torch.float64) * 3).cuda()
    cuts = torch.linspace(-4, 4, 14, dtype=torch.float64)
    lab = (torch.rand(n, generator=g) < 0.5).to(torch.uint8).cuda()
    flat = ext.bucketize_label_counts([xd.contiguous()], [cuts], lab.contiguous(), [16]).cpu()
    ref_bins = torch.bucketize(xd.cpu(), cuts, right=False) + 1  # 1..15
    slots = (ref_bins + 1).clamp(max=15)
    tot_ref = torch.bincount(slots, minlength=16)
    evt_ref = torch.bincount(slots, weights=lab.cpu().double(), minlength=16)
    assert torch.equal(flat[:16], tot_ref.to(torch.int64))
    assert torch.equal(flat[16:32].double(), evt_ref)


@requires_gpu
def test_label_counts_multi_empty_dictionary(ext):
    """Categorical column with an empty dictionary (all nulls)."""
    n = 100_000
    codes = torch.full((n,), -1, dtype=torch.int32).cuda()
    lab = torch.zeros(n, dtype=torch.uint8).cuda()
    flat = ext.label_counts_multi([codes], lab, [1]).cpu()
    assert int(flat[0]) == n and int(flat[1]) == 0
