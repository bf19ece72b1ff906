// Torch bindings for the anovos_amd MI355X kernels (anovos_kernels.hip).
// Compiled by torch.utils.cpp_extension with hipcc for gfx950; streams come
// from the caller's current torch HIP stream so kernels compose with the
// engine's side-stream overlap (core/dist.py SideStream).

#include <torch/extension.h>
#include <c10/hip/HIPStream.h>
#include <hip/hip_runtime.h>

#include <vector>
#include <cstdlib>

extern "C" {
int anovos_moments(const void *const *cols, const int64_t *lens,
                   const double *shifts, int ncols, int nchunks, int dtype,
                   double *partials, double *out, hipStream_t stream);
int anovos_hist(const void *const *cols, const int64_t *lens, int ncols,
                const double *lo, const double *hi, int nbins, int nchunks,
                int dtype, uint64_t *out, hipStream_t stream);
int anovos_bracket_hist(const void *const *cols, const int64_t *lens,
                        const int64_t *colidx, int nbrackets, const double *lo,
                        const double *hi, int nbins, int nchunks, int dtype,
                        uint64_t *out, hipStream_t stream);
int anovos_bucketize(const void *const *cols, const int64_t *lens, int ncols,
                     const double *cutflat, const int64_t *cutoff_off,
                     const int *cutoff_len, int max_ncut, int nchunks, int dtype,
                     int32_t *const *outs, hipStream_t stream);
int anovos_code_counts(const int32_t *codes, int64_t n, int size, int nchunks,
                       uint64_t *out, hipStream_t stream);
int anovos_hll(const void *x, int64_t n, int p, int nchunks, int dtype,
               int32_t *regs, hipStream_t stream);
int anovos_row_null(const void *const *cols, int ncols, int64_t n, int dtype,
                    int32_t *out, hipStream_t stream);
int anovos_hll_multi(const void *const *cols, const int64_t *lens, int ncols,
                     int p, int nchunks, int dtype, int32_t *regs,
                     hipStream_t stream);
int anovos_axpb(const void *const *cols, const int64_t *lens, int ncols,
                const double *a, const double *b, int nchunks, int dtype,
                float *const *outs, hipStream_t stream);
int anovos_fillnan(const void *const *cols, const int64_t *lens, int ncols,
                   const double *fill, int nchunks, int dtype,
                   void *const *outs, hipStream_t stream);
int anovos_bracket_hist_grouped(const void *const *cols, const int64_t *lens,
                                const int *bstart, int ncols, const double *lo,
                                const double *hi, const double *p1lo,
                                const double *p1scale, int p1bins, int nchunks,
                                int dtype, uint64_t *out, hipStream_t stream);
int anovos_bucketize_float(const void *const *cols, const int64_t *lens, int ncols,
                           const double *cutflat, const int64_t *cutoff_off,
                           const int *cutoff_len, int max_ncut, int nchunks, int dtype,
                           float *const *outs, hipStream_t stream);
int anovos_lut_apply_f32(const int32_t *const *cols, const int64_t *lens, int ncols,
                         const float *lutflat, const int64_t *lut_off, int nchunks,
                         float *const *outs, hipStream_t stream);
int anovos_lut_apply_i32(const int32_t *const *cols, const int64_t *lens, int ncols,
                         const int32_t *lutflat, const int64_t *lut_off, int nchunks,
                         int32_t *const *outs, hipStream_t stream);
int anovos_fill_code(const int32_t *const *cols, const int64_t *lens, int ncols,
                     const int32_t *fill, int nchunks, int32_t *const *outs,
                     hipStream_t stream);
int anovos_code_counts_multi(const int32_t *const *cols, const int64_t *lens,
                             const int64_t *offs, const int *sizes, int ncols,
                             int max_slots, int nchunks, uint64_t *out,
                             hipStream_t stream);
int anovos_outlier_clamp(const void *const *cols, const int64_t *lens,
                         int ncols, const double *lo, const double *hi,
                         int nchunks, int mode, int dtype, void *const *outs,
                         uint64_t *counts, hipStream_t stream);
int anovos_moments_hll(const void *const *cols, const int64_t *lens,
                       const double *shifts, int ncols, int p, int nchunks,
                       int dtype, double *partials, double *mom_out,
                       int32_t *regs, hipStream_t stream);
int anovos_centered_gram(const void *const *cols, int64_t n, int k,
                         const float *means, const int *pair_i,
                         const int *pair_j, int npairs, int row_chunks,
                         float *partials, float *gram, hipStream_t stream);
int anovos_centered_gram_sr(const void *const *cols, int64_t n, int k,
                            const float *means, const int *pair_i,
                            const int *pair_j, int npairs, int row_chunks,
                            float *partials, float *gram, hipStream_t stream);
int anovos_gram_sr_grid(int k);
int anovos_bucketize_label_counts(const void *const *cols, const uint8_t *label,
                                  const int64_t *lens, const double *cutflat,
                                  const int64_t *cutoff_off, const int *cutoff_len,
                                  const int64_t *offs, const int *sizes, int ncols,
                                  int max_ncut, int max_slots, int nchunks,
                                  int dtype, uint64_t *out, hipStream_t stream);
int anovos_label_counts_multi(const void *const *cols, const uint8_t *label,
                              const int64_t *lens, const int64_t *offs,
                              const int *sizes, const int *dtypes, int ncols,
                              int max_slots, int nchunks, uint64_t *out,
                              hipStream_t stream);
}

namespace {

hipStream_t current_stream() { return c10::hip::getCurrentHIPStream().stream(); }

void check_hip(int err, const char *what) {
  TORCH_CHECK(err == 0, what, " failed: ", hipGetErrorString((hipError_t)err));
}

int pick_chunks(int64_t n, int ncols) {
  // fill 256 CUs x 8 XCDs: target >= ~2048 workgroups, >= ~64K elems/chunk
  int64_t by_size = (n + (1 << 16) - 1) >> 16;
  int64_t cap = std::max<int64_t>(1, 16384 / std::max(1, ncols));
  int64_t c = std::min<int64_t>(std::max<int64_t>(by_size, 1), cap);
  int64_t floor_c = std::min<int64_t>(by_size, std::max<int64_t>(1, 2048 / std::max(1, ncols)));
  return (int)std::max<int64_t>(c, std::max<int64_t>(floor_c, 1));
}

struct PtrPack {
  torch::Tensor dev;  // int64 tensor of device pointers
};

// Upload an array of device pointers / lengths to the GPU.
torch::Tensor to_device_i64(const std::vector<int64_t> &vals, const torch::Device &dev) {
  auto cpu = torch::from_blob((void *)vals.data(), {(int64_t)vals.size()},
                              torch::TensorOptions().dtype(torch::kInt64))
                 .clone();
  return cpu.to(dev, /*non_blocking=*/true);
}

torch::Tensor to_device_f64(const std::vector<double> &vals, const torch::Device &dev) {
  auto cpu = torch::from_blob((void *)vals.data(), {(int64_t)vals.size()},
                              torch::TensorOptions().dtype(torch::kFloat64))
                 .clone();
  return cpu.to(dev, /*non_blocking=*/true);
}

int dtype_code(const torch::Tensor &t) {
  if (t.scalar_type() == torch::kFloat32) return 0;
  if (t.scalar_type() == torch::kFloat64) return 1;
  TORCH_CHECK(false, "expected float32/float64 column, got ", t.scalar_type());
  return -1;
}

}  // namespace

torch::Tensor column_moments(std::vector<torch::Tensor> cols, std::vector<double> shifts) {
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(shifts.empty() || shifts.size() == cols.size(), "shifts must match cols");
  auto device = cols[0].device();
  auto out = torch::zeros({(int64_t)cols.size(), 9},
                          torch::TensorOptions().dtype(torch::kFloat64).device(device));
  // group by dtype, one fused launch per dtype
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, idx;
    std::vector<double> sh;
    for (size_t i = 0; i < cols.size(); ++i) {
      auto &t = cols[i];
      TORCH_CHECK(t.is_contiguous() && t.device() == device, "columns must be contiguous, same device");
      if (dtype_code(t) != pass) continue;
      ptrs.push_back((int64_t)t.data_ptr());
      lens.push_back(t.numel());
      sh.push_back(shifts.empty() ? 0.0 : shifts[i]);
      idx.push_back((int64_t)i);
    }
    if (ptrs.empty()) continue;
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto dsh = to_device_f64(sh, device);
    auto partials = torch::empty({(int64_t)ncols * nchunks, 9},
                                 torch::TensorOptions().dtype(torch::kFloat64).device(device));
    auto sub = torch::empty({ncols, 9}, torch::TensorOptions().dtype(torch::kFloat64).device(device));
    check_hip(anovos_moments((const void *const *)dptr.data_ptr<int64_t>(),
                             dlen.data_ptr<int64_t>(), dsh.data_ptr<double>(),
                             ncols, nchunks, pass,
                             partials.data_ptr<double>(), sub.data_ptr<double>(),
                             current_stream()),
              "anovos_moments");
    auto didx = to_device_i64(idx, device);
    out.index_copy_(0, didx, sub);
  }
  return out;
}

torch::Tensor column_histograms(std::vector<torch::Tensor> cols, torch::Tensor lo,
                                torch::Tensor hi, int64_t nbins) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  auto lo_d = lo.to(torch::kFloat64).to(device);
  auto hi_d = hi.to(torch::kFloat64).to(device);
  auto out = torch::zeros({(int64_t)cols.size(), nbins},
                          torch::TensorOptions().dtype(torch::kInt64).device(device));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, idx;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      idx.push_back((int64_t)i);
    }
    if (ptrs.empty()) continue;
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto didx_cpu = idx;
    auto sub = torch::zeros({ncols, nbins}, torch::TensorOptions().dtype(torch::kInt64).device(device));
    auto lo_sel = lo_d.index_select(0, to_device_i64(didx_cpu, device));
    auto hi_sel = hi_d.index_select(0, to_device_i64(didx_cpu, device));
    check_hip(anovos_hist((const void *const *)dptr.data_ptr<int64_t>(),
                          dlen.data_ptr<int64_t>(), ncols, lo_sel.data_ptr<double>(),
                          hi_sel.data_ptr<double>(), (int)nbins, nchunks, pass,
                          (uint64_t *)sub.data_ptr<int64_t>(), current_stream()),
              "anovos_hist");
    out.index_copy_(0, to_device_i64(didx_cpu, device), sub);
  }
  return out;
}

torch::Tensor bracket_histograms(std::vector<torch::Tensor> cols, torch::Tensor colidx,
                                 torch::Tensor lo, torch::Tensor hi, int64_t nbins) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  int dtype = dtype_code(cols[0]);
  for (auto &t : cols) TORCH_CHECK(dtype_code(t) == dtype, "bracket_histograms: mixed dtypes unsupported");
  std::vector<int64_t> ptrs, lens;
  for (auto &t : cols) {
    ptrs.push_back((int64_t)t.data_ptr());
    lens.push_back(t.numel());
  }
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto ci = colidx.to(torch::kInt64).to(device);
  auto lo_d = lo.to(torch::kFloat64).to(device);
  auto hi_d = hi.to(torch::kFloat64).to(device);
  int nbrackets = (int)ci.numel();
  int64_t maxn = *std::max_element(lens.begin(), lens.end());
  int nchunks = pick_chunks(maxn, nbrackets);
  auto out = torch::zeros({nbrackets, nbins}, torch::TensorOptions().dtype(torch::kInt64).device(device));
  check_hip(anovos_bracket_hist((const void *const *)dptr.data_ptr<int64_t>(),
                                dlen.data_ptr<int64_t>(), ci.data_ptr<int64_t>(),
                                nbrackets, lo_d.data_ptr<double>(), hi_d.data_ptr<double>(),
                                (int)nbins, nchunks, dtype,
                                (uint64_t *)out.data_ptr<int64_t>(), current_stream()),
            "anovos_bracket_hist");
  return out;
}

std::vector<torch::Tensor> bucketize_columns(std::vector<torch::Tensor> cols,
                                             std::vector<torch::Tensor> cutoffs) {
  TORCH_CHECK(cols.size() == cutoffs.size(), "cols/cutoffs size mismatch");
  auto device = cols[0].device();
  std::vector<torch::Tensor> outs;
  for (auto &t : cols)
    outs.push_back(torch::empty_like(t, torch::TensorOptions().dtype(torch::kInt32).device(device)));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, optrs, offs;
    std::vector<double> flat;
    std::vector<int> clens;
    int max_ncut = 1;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      optrs.push_back((int64_t)outs[i].data_ptr());
      auto cc = cutoffs[i].to(torch::kFloat64).cpu().contiguous();
      offs.push_back((int64_t)flat.size());
      const double *cd = cc.data_ptr<double>();
      flat.insert(flat.end(), cd, cd + cc.numel());
      clens.push_back((int)cc.numel());
      max_ncut = std::max(max_ncut, (int)cc.numel());
    }
    if (ptrs.empty()) continue;
    if (flat.empty()) flat.push_back(0.0);  // from_blob on empty data() segfaults
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto dout = to_device_i64(optrs, device);
    auto doff = to_device_i64(offs, device);
    auto dflat = torch::from_blob(flat.data(), {(int64_t)std::max<size_t>(flat.size(), 1)},
                                  torch::TensorOptions().dtype(torch::kFloat64))
                     .clone()
                     .to(device);
    auto dclen = torch::from_blob(clens.data(), {(int64_t)clens.size()},
                                  torch::TensorOptions().dtype(torch::kInt32))
                     .clone()
                     .to(device);
    check_hip(anovos_bucketize((const void *const *)dptr.data_ptr<int64_t>(),
                               dlen.data_ptr<int64_t>(), ncols, dflat.data_ptr<double>(),
                               doff.data_ptr<int64_t>(), dclen.data_ptr<int>(), max_ncut,
                               nchunks, pass, (int32_t *const *)dout.data_ptr<int64_t>(),
                               current_stream()),
              "anovos_bucketize");
  }
  return outs;
}

torch::Tensor code_counts(torch::Tensor codes, int64_t size) {
  TORCH_CHECK(codes.scalar_type() == torch::kInt32, "codes must be int32");
  auto device = codes.device();
  auto out = torch::zeros({std::max<int64_t>(size, 1)},
                          torch::TensorOptions().dtype(torch::kInt64).device(device));
  if (size == 0 || codes.numel() == 0) return out.narrow(0, 0, size);
  int nchunks = pick_chunks(codes.numel(), 1);
  nchunks = std::min(nchunks, 8192);
  check_hip(anovos_code_counts(codes.data_ptr<int32_t>(), codes.numel(), (int)size,
                               nchunks, (uint64_t *)out.data_ptr<int64_t>(),
                               current_stream()),
            "anovos_code_counts");
  return out;
}

torch::Tensor hll_registers(torch::Tensor x, int64_t p) {
  auto device = x.device();
  auto regs = torch::zeros({(int64_t)1 << p},
                           torch::TensorOptions().dtype(torch::kInt32).device(device));
  if (x.numel() == 0) return regs;
  int nchunks = std::min(pick_chunks(x.numel(), 1), 4096);
  check_hip(anovos_hll(x.data_ptr(), x.numel(), (int)p, nchunks, dtype_code(x),
                       regs.data_ptr<int32_t>(), current_stream()),
            "anovos_hll");
  return regs;
}

void row_null_counts_num(std::vector<torch::Tensor> cols, torch::Tensor out) {
  // accepts float32/float64 (NaN null) AND int32 dictionary codes (-1 null)
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(out.scalar_type() == torch::kInt32, "out must be int32");
  auto device = cols[0].device();
  for (int pass = 0; pass < 3; ++pass) {
    std::vector<int64_t> ptrs;
    for (auto &t : cols) {
      int dc = (t.scalar_type() == torch::kInt32) ? 2 : dtype_code(t);
      if (dc != pass) continue;
      TORCH_CHECK(t.numel() == out.numel(), "column length mismatch");
      TORCH_CHECK(t.is_contiguous(), "columns must be contiguous");
      ptrs.push_back((int64_t)t.data_ptr());
    }
    if (ptrs.empty()) continue;
    auto dptr = to_device_i64(ptrs, device);
    check_hip(anovos_row_null((const void *const *)dptr.data_ptr<int64_t>(),
                              (int)ptrs.size(), out.numel(), pass,
                              out.data_ptr<int32_t>(), current_stream()),
              "anovos_row_null");
  }
}

torch::Tensor hll_registers_multi(std::vector<torch::Tensor> cols, int64_t p) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  auto regs = torch::zeros({(int64_t)cols.size(), (int64_t)1 << p},
                           torch::TensorOptions().dtype(torch::kInt32).device(device));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, rptrs, idx;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      idx.push_back((int64_t)i);
    }
    if (ptrs.empty()) continue;
    // contiguous sub-block trick: launch on a per-pass register buffer
    auto sub = torch::zeros({(int64_t)ptrs.size(), (int64_t)1 << p},
                            torch::TensorOptions().dtype(torch::kInt32).device(device));
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = std::max(1, std::min((int)((maxn + (1 << 20) - 1) >> 20), (int)(4096 / std::max<size_t>(ptrs.size(), 1) + 1)));
    check_hip(anovos_hll_multi((const void *const *)dptr.data_ptr<int64_t>(),
                               dlen.data_ptr<int64_t>(), (int)ptrs.size(), (int)p,
                               nchunks, pass, sub.data_ptr<int32_t>(), current_stream()),
              "anovos_hll_multi");
    regs.index_copy_(0, to_device_i64(idx, device), sub);
  }
  return regs;
}

std::vector<torch::Tensor> scale_columns(std::vector<torch::Tensor> cols,
                                         torch::Tensor a, torch::Tensor b) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  auto a_c = a.to(torch::kFloat64).cpu().contiguous();
  auto b_c = b.to(torch::kFloat64).cpu().contiguous();
  std::vector<torch::Tensor> outs;
  for (auto &t : cols)
    outs.push_back(torch::empty(t.sizes(), torch::TensorOptions().dtype(torch::kFloat32).device(device)));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, optrs;
    std::vector<double> av, bv;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      optrs.push_back((int64_t)outs[i].data_ptr());
      av.push_back(a_c.data_ptr<double>()[i]);
      bv.push_back(b_c.data_ptr<double>()[i]);
    }
    if (ptrs.empty()) continue;
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, (int)ptrs.size());
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto dout = to_device_i64(optrs, device);
    auto da = torch::from_blob(av.data(), {(int64_t)av.size()}, torch::TensorOptions().dtype(torch::kFloat64)).clone().to(device);
    auto db = torch::from_blob(bv.data(), {(int64_t)bv.size()}, torch::TensorOptions().dtype(torch::kFloat64)).clone().to(device);
    check_hip(anovos_axpb((const void *const *)dptr.data_ptr<int64_t>(),
                          dlen.data_ptr<int64_t>(), (int)ptrs.size(),
                          da.data_ptr<double>(), db.data_ptr<double>(), nchunks, pass,
                          (float *const *)dout.data_ptr<int64_t>(), current_stream()),
              "anovos_axpb");
  }
  return outs;
}

std::vector<torch::Tensor> fill_nan_columns(std::vector<torch::Tensor> cols, torch::Tensor fill) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  auto f_c = fill.to(torch::kFloat64).cpu().contiguous();
  std::vector<torch::Tensor> outs;
  for (auto &t : cols) outs.push_back(torch::empty_like(t));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, optrs;
    std::vector<double> fv;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      optrs.push_back((int64_t)outs[i].data_ptr());
      fv.push_back(f_c.data_ptr<double>()[i]);
    }
    if (ptrs.empty()) continue;
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, (int)ptrs.size());
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto dout = to_device_i64(optrs, device);
    auto df = torch::from_blob(fv.data(), {(int64_t)fv.size()}, torch::TensorOptions().dtype(torch::kFloat64)).clone().to(device);
    check_hip(anovos_fillnan((const void *const *)dptr.data_ptr<int64_t>(),
                             dlen.data_ptr<int64_t>(), (int)ptrs.size(),
                             df.data_ptr<double>(), nchunks, pass,
                             (void *const *)dout.data_ptr<int64_t>(), current_stream()),
              "anovos_fillnan");
  }
  return outs;
}

// Fused categorical null-fill: out_i = (code == -1) ? fill_i : code.
std::vector<torch::Tensor> fill_code_columns(std::vector<torch::Tensor> cols,
                                             std::vector<int64_t> fills) {
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(fills.size() == cols.size(), "fills must match cols");
  auto device = cols[0].device();
  std::vector<torch::Tensor> outs;
  std::vector<int64_t> ptrs, lens, optrs;
  std::vector<int64_t> fv;
  for (size_t i = 0; i < cols.size(); ++i) {
    auto &t = cols[i];
    TORCH_CHECK(t.scalar_type() == torch::kInt32 && t.is_contiguous() && t.device() == device,
                "expected contiguous int32 code columns on one device");
    outs.push_back(torch::empty_like(t));
    ptrs.push_back((int64_t)t.data_ptr());
    lens.push_back(t.numel());
    optrs.push_back((int64_t)outs.back().data_ptr());
    fv.push_back(fills[i]);
  }
  int64_t maxn = *std::max_element(lens.begin(), lens.end());
  int nchunks = pick_chunks(maxn, (int)ptrs.size());
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto dout = to_device_i64(optrs, device);
  auto df64 = to_device_i64(fv, device).to(torch::kInt32);
  check_hip(anovos_fill_code((const int32_t *const *)dptr.data_ptr<int64_t>(),
                             dlen.data_ptr<int64_t>(), (int)ptrs.size(),
                             df64.data_ptr<int32_t>(), nchunks,
                             (int32_t *const *)dout.data_ptr<int64_t>(), current_stream()),
            "anovos_fill_code");
  return outs;
}

torch::Tensor bracket_histograms_grouped(std::vector<torch::Tensor> cols,
                                         torch::Tensor bracket_col,
                                         torch::Tensor lo, torch::Tensor hi,
                                         torch::Tensor p1lo, torch::Tensor p1scale,
                                         int64_t p1bins) {
  // brackets MUST be sorted by column index; returns [nbrackets, 512].
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  auto bc = bracket_col.to(torch::kInt64).cpu().contiguous();
  // dtype uniformity only matters for REFERENCED columns (the host
  // splits mixed-dtype bracket sets into per-dtype launches; columns
  // without brackets in this launch are never read)
  int dtype = -1;
  {
    const int64_t *b = bc.data_ptr<int64_t>();
    for (int i = 0; i < (int)bc.numel(); ++i) {
      TORCH_CHECK(b[i] >= 0 && b[i] < (int64_t)cols.size(), "bad bracket col");
      int d = dtype_code(cols[b[i]]);
      if (dtype < 0) dtype = d;
      TORCH_CHECK(d == dtype, "grouped brackets: mixed dtypes in one launch");
    }
    if (dtype < 0) dtype = dtype_code(cols[0]);
  }
  int ncols = (int)cols.size();
  int nb = (int)bc.numel();
  std::vector<int> bstart(ncols + 1, 0);
  {
    const int64_t *b = bc.data_ptr<int64_t>();
    for (int i = 0; i < nb; ++i) {
      TORCH_CHECK(b[i] >= 0 && b[i] < ncols, "bad bracket col");
      if (i) TORCH_CHECK(b[i] >= b[i - 1], "brackets must be sorted by column");
      bstart[b[i] + 1]++;
    }
    for (int c = 0; c < ncols; ++c) {
      TORCH_CHECK(bstart[c + 1] <= 16, "at most 16 brackets per column");
      bstart[c + 1] += bstart[c];
    }
  }
  std::vector<int64_t> ptrs, lens;
  for (auto &t : cols) {
    ptrs.push_back((int64_t)t.data_ptr());
    lens.push_back(t.numel());
  }
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto dbs = torch::from_blob(bstart.data(), {(int64_t)bstart.size()}, torch::TensorOptions().dtype(torch::kInt32)).clone().to(device);
  auto lo_d = lo.to(torch::kFloat64).to(device);
  auto hi_d = hi.to(torch::kFloat64).to(device);
  auto p1lo_d = p1lo.to(torch::kFloat64).to(device).contiguous();
  auto p1scale_d = p1scale.to(torch::kFloat64).to(device).contiguous();
  int64_t maxn = *std::max_element(lens.begin(), lens.end());
  int nchunks = pick_chunks(maxn, ncols);
  auto out = torch::zeros({nb, 512}, torch::TensorOptions().dtype(torch::kInt64).device(device));
  check_hip(anovos_bracket_hist_grouped((const void *const *)dptr.data_ptr<int64_t>(),
                                        dlen.data_ptr<int64_t>(), dbs.data_ptr<int>(), ncols,
                                        lo_d.data_ptr<double>(), hi_d.data_ptr<double>(),
                                        p1lo_d.data_ptr<double>(), p1scale_d.data_ptr<double>(),
                                        (int)p1bins, nchunks, dtype,
                                        (uint64_t *)out.data_ptr<int64_t>(),
                                        current_stream()),
            "anovos_bracket_hist_grouped");
  return out;
}

std::vector<torch::Tensor> bucketize_columns_float(std::vector<torch::Tensor> cols,
                                                   std::vector<torch::Tensor> cutoffs) {
  TORCH_CHECK(cols.size() == cutoffs.size(), "cols/cutoffs size mismatch");
  auto device = cols[0].device();
  std::vector<torch::Tensor> outs;
  for (auto &t : cols)
    outs.push_back(torch::empty(t.sizes(), torch::TensorOptions().dtype(torch::kFloat32).device(device)));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, optrs, offs;
    std::vector<double> flat;
    std::vector<int> clens;
    int max_ncut = 1;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      optrs.push_back((int64_t)outs[i].data_ptr());
      auto cc = cutoffs[i].to(torch::kFloat64).cpu().contiguous();
      offs.push_back((int64_t)flat.size());
      const double *cd = cc.data_ptr<double>();
      flat.insert(flat.end(), cd, cd + cc.numel());
      clens.push_back((int)cc.numel());
      max_ncut = std::max(max_ncut, (int)cc.numel());
    }
    if (ptrs.empty()) continue;
    if (flat.empty()) flat.push_back(0.0);  // from_blob on empty data() segfaults
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto dout = to_device_i64(optrs, device);
    auto doff = to_device_i64(offs, device);
    auto dflat = torch::from_blob(flat.data(), {(int64_t)std::max<size_t>(flat.size(), 1)},
                                  torch::TensorOptions().dtype(torch::kFloat64)).clone().to(device);
    auto dclen = torch::from_blob(clens.data(), {(int64_t)clens.size()},
                                  torch::TensorOptions().dtype(torch::kInt32)).clone().to(device);
    check_hip(anovos_bucketize_float((const void *const *)dptr.data_ptr<int64_t>(),
                                     dlen.data_ptr<int64_t>(), ncols, dflat.data_ptr<double>(),
                                     doff.data_ptr<int64_t>(), dclen.data_ptr<int>(), max_ncut,
                                     nchunks, pass, (float *const *)dout.data_ptr<int64_t>(),
                                     current_stream()),
              "anovos_bucketize_float");
  }
  return outs;
}

std::vector<torch::Tensor> lut_apply_f32(std::vector<torch::Tensor> cols,
                                         std::vector<torch::Tensor> luts) {
  TORCH_CHECK(cols.size() == luts.size(), "cols/luts size mismatch");
  auto device = cols[0].device();
  std::vector<int64_t> ptrs, lens, optrs, offs;
  std::vector<float> flat;
  std::vector<torch::Tensor> outs;
  for (size_t i = 0; i < cols.size(); ++i) {
    TORCH_CHECK(cols[i].scalar_type() == torch::kInt32, "codes must be int32");
    outs.push_back(torch::empty(cols[i].sizes(), torch::TensorOptions().dtype(torch::kFloat32).device(device)));
    ptrs.push_back((int64_t)cols[i].data_ptr());
    lens.push_back(cols[i].numel());
    optrs.push_back((int64_t)outs[i].data_ptr());
    auto lc = luts[i].to(torch::kFloat32).cpu().contiguous();
    offs.push_back((int64_t)flat.size());
    const float *ld = lc.data_ptr<float>();
    flat.insert(flat.end(), ld, ld + lc.numel());
  }
  int64_t maxn = *std::max_element(lens.begin(), lens.end());
  int nchunks = pick_chunks(maxn, (int)cols.size());
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto dout = to_device_i64(optrs, device);
  auto doff = to_device_i64(offs, device);
  auto dflat = torch::from_blob(flat.data(), {(int64_t)std::max<size_t>(flat.size(), 1)},
                                torch::TensorOptions().dtype(torch::kFloat32)).clone().to(device);
  check_hip(anovos_lut_apply_f32((const int32_t *const *)dptr.data_ptr<int64_t>(),
                                 dlen.data_ptr<int64_t>(), (int)cols.size(),
                                 dflat.data_ptr<float>(), doff.data_ptr<int64_t>(), nchunks,
                                 (float *const *)dout.data_ptr<int64_t>(), current_stream()),
            "anovos_lut_apply_f32");
  return outs;
}

std::vector<torch::Tensor> lut_apply_i32(std::vector<torch::Tensor> cols,
                                         std::vector<torch::Tensor> luts) {
  TORCH_CHECK(cols.size() == luts.size(), "cols/luts size mismatch");
  auto device = cols[0].device();
  std::vector<int64_t> ptrs, lens, optrs, offs;
  std::vector<int32_t> flat;
  std::vector<torch::Tensor> outs;
  for (size_t i = 0; i < cols.size(); ++i) {
    TORCH_CHECK(cols[i].scalar_type() == torch::kInt32, "codes must be int32");
    outs.push_back(torch::empty(cols[i].sizes(), torch::TensorOptions().dtype(torch::kInt32).device(device)));
    ptrs.push_back((int64_t)cols[i].data_ptr());
    lens.push_back(cols[i].numel());
    optrs.push_back((int64_t)outs[i].data_ptr());
    auto lc = luts[i].to(torch::kInt32).cpu().contiguous();
    offs.push_back((int64_t)flat.size());
    const int32_t *ld = lc.data_ptr<int32_t>();
    flat.insert(flat.end(), ld, ld + lc.numel());
  }
  int64_t maxn = *std::max_element(lens.begin(), lens.end());
  int nchunks = pick_chunks(maxn, (int)cols.size());
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto dout = to_device_i64(optrs, device);
  auto doff = to_device_i64(offs, device);
  auto dflat = torch::from_blob(flat.data(), {(int64_t)std::max<size_t>(flat.size(), 1)},
                                torch::TensorOptions().dtype(torch::kInt32)).clone().to(device);
  check_hip(anovos_lut_apply_i32((const int32_t *const *)dptr.data_ptr<int64_t>(),
                                 dlen.data_ptr<int64_t>(), (int)cols.size(),
                                 dflat.data_ptr<int32_t>(), doff.data_ptr<int64_t>(), nchunks,
                                 (int32_t *const *)dout.data_ptr<int64_t>(), current_stream()),
            "anovos_lut_apply_i32");
  return outs;
}


// K8: bf16 MFMA centered Gram — gram[i][j] = sum (x_i - mean_i)(x_j - mean_j)
torch::Tensor centered_gram_bf16(std::vector<torch::Tensor> cols, torch::Tensor means) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  const int k = (int)cols.size();
  std::vector<int64_t> ptrs;
  int64_t n = cols[0].numel();
  for (auto &t : cols) {
    TORCH_CHECK(t.is_contiguous() && t.device() == device, "columns must be contiguous, same device");
    TORCH_CHECK(t.scalar_type() == torch::kFloat32, "centered_gram_bf16 expects float32 columns");
    TORCH_CHECK(t.numel() == n, "all columns must have equal length");
    ptrs.push_back((int64_t)t.data_ptr());
  }
  auto means_d = means.to(device, torch::kFloat32).contiguous();
  TORCH_CHECK(means_d.numel() == k, "means length mismatch");
  const int kt = (k + 15) / 16;
  std::vector<int64_t> pi64, pj64;
  for (int i = 0; i < kt; ++i)
    for (int j = i; j < kt; ++j) {
      pi64.push_back(i);
      pj64.push_back(j);
    }
  const int npairs = (int)pi64.size();
  auto pi = to_device_i64(pi64, device).to(torch::kInt32);
  auto pj = to_device_i64(pj64, device).to(torch::kInt32);
  auto dptr = to_device_i64(ptrs, device);
  auto gram = torch::zeros({k, k}, torch::TensorOptions().dtype(torch::kFloat32).device(device));
  if (kt <= 13) {
    // single-read kernel: one block stages a 32-row slab of ALL columns
    // through LDS; HBM traffic = n*k*4 bytes (vs ~kt x for pair-parallel)
    const int64_t steps_total = (n + 31) / 32;
    // one full wave of resident blocks (occupancy x CUs): maximizes each
    // block's contiguous per-column streams (see anovos_gram_sr_grid)
    int64_t target_chunks = anovos_gram_sr_grid(k);
    if (const char *e = getenv("ANOVOS_GRAM_CHUNKS")) target_chunks = atoll(e);
    int row_chunks = (int)std::min<int64_t>(target_chunks, std::max<int64_t>(1, steps_total));
    auto partials = torch::empty({(int64_t)row_chunks * npairs, 256},
                                 torch::TensorOptions().dtype(torch::kFloat32).device(device));
    check_hip(anovos_centered_gram_sr((const void *const *)dptr.data_ptr<int64_t>(), n, k,
                                      means_d.data_ptr<float>(), pi.data_ptr<int>(),
                                      pj.data_ptr<int>(), npairs, row_chunks,
                                      partials.data_ptr<float>(), gram.data_ptr<float>(),
                                      current_stream()),
              "anovos_centered_gram_sr");
    return gram;
  }
  // wide matrices: pair-parallel kernel (re-reads columns, but register
  // budget no longer admits the all-pairs accumulator set)
  int row_chunks = (int)std::min<int64_t>(std::max<int64_t>(1, 2048 / std::max(npairs, 1)),
                                          std::max<int64_t>(1, n >> 16));
  row_chunks = std::max(row_chunks, 1);
  auto partials = torch::empty({(int64_t)npairs * row_chunks, 256},
                               torch::TensorOptions().dtype(torch::kFloat32).device(device));
  check_hip(anovos_centered_gram((const void *const *)dptr.data_ptr<int64_t>(), n, k,
                                 means_d.data_ptr<float>(), pi.data_ptr<int>(),
                                 pj.data_ptr<int>(), npairs, row_chunks,
                                 partials.data_ptr<float>(), gram.data_ptr<float>(),
                                 current_stream()),
            "anovos_centered_gram");
  return gram;
}


// K5 fused: multi-column code counts (+ null slot per column).
// Returns flat int64 tensor of length sum(sizes_i + 1).
torch::Tensor code_counts_multi(std::vector<torch::Tensor> cols, std::vector<int64_t> sizes) {
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(cols.size() == sizes.size(), "sizes mismatch");
  auto device = cols[0].device();
  std::vector<int64_t> ptrs, lens, offs;
  std::vector<int64_t> sz64;
  int64_t off = 0, maxn = 0;
  int max_slots = 0;
  for (size_t i = 0; i < cols.size(); ++i) {
    auto &t = cols[i];
    TORCH_CHECK(t.is_contiguous() && t.scalar_type() == torch::kInt32, "int32 code columns required");
    ptrs.push_back((int64_t)t.data_ptr());
    lens.push_back(t.numel());
    maxn = std::max(maxn, t.numel());
    offs.push_back(off);
    off += sizes[i] + 1;
    sz64.push_back(sizes[i]);
    max_slots = std::max(max_slots, (int)sizes[i] + 1);
  }
  int ncols = (int)cols.size();
  int nchunks = pick_chunks(maxn, ncols);
  auto out = torch::zeros({off}, torch::TensorOptions().dtype(torch::kInt64).device(device));
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto doff = to_device_i64(offs, device);
  auto dsz = to_device_i64(sz64, device).to(torch::kInt32);
  check_hip(anovos_code_counts_multi((const int32_t *const *)dptr.data_ptr<int64_t>(),
                                     dlen.data_ptr<int64_t>(), doff.data_ptr<int64_t>(),
                                     dsz.data_ptr<int>(), ncols, max_slots, nchunks,
                                     (uint64_t *)out.data_ptr<int64_t>(), current_stream()),
            "anovos_code_counts_multi");
  return out;
}

// K9 fused: label-conditioned histograms, ALL columns in one launch.
// Per column at out[off]: [slots] totals then [slots] event counts.
// Column dtype drives the slot mapping (f32 binned vs int32 codes).
torch::Tensor label_counts_multi(std::vector<torch::Tensor> cols,
                                 torch::Tensor label,
                                 std::vector<int64_t> sizes) {
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(cols.size() == sizes.size(), "sizes mismatch");
  TORCH_CHECK(label.is_contiguous() && label.scalar_type() == torch::kUInt8,
              "uint8 label required");
  auto device = cols[0].device();
  std::vector<int64_t> ptrs, lens, offs, dt, sz;
  int64_t off = 0, maxn = 0;
  int max_slots = 0;
  for (size_t i = 0; i < cols.size(); ++i) {
    auto &t = cols[i];
    TORCH_CHECK(t.is_contiguous(), "contiguous columns required");
    int dtype;
    if (t.scalar_type() == torch::kFloat) dtype = 0;
    else if (t.scalar_type() == torch::kInt) dtype = 1;
    else TORCH_CHECK(false, "f32 or int32 columns required");
    TORCH_CHECK(t.numel() == label.numel(), "column/label length mismatch");
    TORCH_CHECK(sizes[i] >= 1 && sizes[i] <= 8192, "slot count out of LDS range");
    ptrs.push_back((int64_t)t.data_ptr());
    lens.push_back(t.numel());
    offs.push_back(off);
    off += 2 * sizes[i];
    dt.push_back(dtype);
    sz.push_back(sizes[i]);
    maxn = std::max(maxn, t.numel());
    max_slots = std::max(max_slots, (int)sizes[i]);
  }
  int ncols = (int)cols.size();
  int nchunks = pick_chunks(maxn, ncols);
  auto out = torch::zeros({off}, torch::TensorOptions().dtype(torch::kInt64).device(device));
  auto dptr = to_device_i64(ptrs, device);
  auto dlen = to_device_i64(lens, device);
  auto doff = to_device_i64(offs, device);
  auto dsz = to_device_i64(sz, device).to(torch::kInt32);
  auto ddt = to_device_i64(dt, device).to(torch::kInt32);
  check_hip(anovos_label_counts_multi(
                (const void *const *)dptr.data_ptr<int64_t>(),
                label.data_ptr<uint8_t>(), dlen.data_ptr<int64_t>(),
                doff.data_ptr<int64_t>(), dsz.data_ptr<int>(),
                ddt.data_ptr<int>(), ncols, max_slots, nchunks,
                (uint64_t *)out.data_ptr<int64_t>(), current_stream()),
            "anovos_label_counts_multi");
  return out;
}

// K6+K9 fused: bucketize against per-column cutoffs + label-conditioned
// counts in one read of the RAW numeric columns (no binned
// materialization). Returns flat int64: per column [slots] totals then
// [slots] events at offsets in the ORIGINAL column order.
torch::Tensor bucketize_label_counts(std::vector<torch::Tensor> cols,
                                     std::vector<torch::Tensor> cutoffs,
                                     torch::Tensor label,
                                     std::vector<int64_t> sizes) {
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(cols.size() == cutoffs.size() && cols.size() == sizes.size(), "size mismatch");
  TORCH_CHECK(label.is_contiguous() && label.scalar_type() == torch::kUInt8, "uint8 label required");
  auto device = cols[0].device();
  std::vector<int64_t> out_offs(cols.size());
  int64_t total = 0;
  for (size_t i = 0; i < cols.size(); ++i) {
    TORCH_CHECK(sizes[i] >= 1 && sizes[i] <= 8192, "slot count out of LDS range");
    TORCH_CHECK(cols[i].numel() == label.numel(), "column/label length mismatch");
    out_offs[i] = total;
    total += 2 * sizes[i];
  }
  auto out = torch::zeros({total}, torch::TensorOptions().dtype(torch::kInt64).device(device));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, offs, coffs;
    std::vector<double> flat;
    std::vector<int> clens;
    std::vector<int64_t> sz;
    int max_ncut = 1, max_slots = 1;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      TORCH_CHECK(cols[i].is_contiguous(), "contiguous columns required");
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      offs.push_back(out_offs[i]);
      auto cc = cutoffs[i].to(torch::kFloat64).cpu().contiguous();
      coffs.push_back((int64_t)flat.size());
      const double *cd = cc.data_ptr<double>();
      flat.insert(flat.end(), cd, cd + cc.numel());
      clens.push_back((int)cc.numel());
      sz.push_back(sizes[i]);
      max_ncut = std::max(max_ncut, (int)cc.numel());
      max_slots = std::max(max_slots, (int)sizes[i]);
    }
    if (ptrs.empty()) continue;
    if (flat.empty()) flat.push_back(0.0);  // from_blob on an empty vector's data() segfaults
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto doff = to_device_i64(offs, device);
    auto dcoff = to_device_i64(coffs, device);
    auto dflat = torch::from_blob(flat.data(), {(int64_t)std::max<size_t>(flat.size(), 1)},
                                  torch::TensorOptions().dtype(torch::kFloat64)).clone().to(device);
    auto dclen = torch::from_blob(clens.data(), {(int64_t)clens.size()},
                                  torch::TensorOptions().dtype(torch::kInt32)).clone().to(device);
    auto dsz = to_device_i64(sz, device).to(torch::kInt32);
    check_hip(anovos_bucketize_label_counts(
                  (const void *const *)dptr.data_ptr<int64_t>(),
                  label.data_ptr<uint8_t>(), dlen.data_ptr<int64_t>(),
                  dflat.data_ptr<double>(), dcoff.data_ptr<int64_t>(),
                  dclen.data_ptr<int>(), doff.data_ptr<int64_t>(),
                  dsz.data_ptr<int>(), ncols, max_ncut, max_slots, nchunks,
                  pass, (uint64_t *)out.data_ptr<int64_t>(), current_stream()),
              "anovos_bucketize_label_counts");
  }
  return out;
}

// K10/K11 fused: outlier counts + clamp/null treatment in one launch.
// mode: 0 count only, 1 clamp to bounds, 2 null-out. Returns
// (counts [k,2] int64, outs list — empty when mode==0).
std::tuple<torch::Tensor, std::vector<torch::Tensor>> outlier_clamp_columns(
    std::vector<torch::Tensor> cols, torch::Tensor lo, torch::Tensor hi, int64_t mode) {
  TORCH_CHECK(!cols.empty(), "no columns");
  auto device = cols[0].device();
  auto lo_d = lo.to(torch::kFloat64).to(device).contiguous();
  auto hi_d = hi.to(torch::kFloat64).to(device).contiguous();
  auto counts = torch::zeros({(int64_t)cols.size(), 2},
                             torch::TensorOptions().dtype(torch::kInt64).device(device));
  std::vector<torch::Tensor> outs_all(cols.size());
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, optrs, idx;
    std::vector<torch::Tensor> outs;
    for (size_t i = 0; i < cols.size(); ++i) {
      if (dtype_code(cols[i]) != pass) continue;
      TORCH_CHECK(cols[i].is_contiguous(), "columns must be contiguous");
      ptrs.push_back((int64_t)cols[i].data_ptr());
      lens.push_back(cols[i].numel());
      idx.push_back((int64_t)i);
      if (mode) {
        auto o = torch::empty_like(cols[i]);
        optrs.push_back((int64_t)o.data_ptr());
        outs.push_back(o);
      }
    }
    if (ptrs.empty()) continue;
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto didx = to_device_i64(idx, device);
    auto lo_sel = lo_d.index_select(0, didx);
    auto hi_sel = hi_d.index_select(0, didx);
    auto sub = torch::zeros({ncols, 2}, torch::TensorOptions().dtype(torch::kInt64).device(device));
    torch::Tensor dout;
    if (mode) dout = to_device_i64(optrs, device);
    check_hip(anovos_outlier_clamp((const void *const *)dptr.data_ptr<int64_t>(),
                                   dlen.data_ptr<int64_t>(), ncols,
                                   lo_sel.data_ptr<double>(), hi_sel.data_ptr<double>(),
                                   nchunks, (int)mode, pass,
                                   mode ? (void *const *)dout.data_ptr<int64_t>() : nullptr,
                                   (uint64_t *)sub.data_ptr<int64_t>(), current_stream()),
              "anovos_outlier_clamp");
    counts.index_copy_(0, didx, sub);
    for (size_t j = 0; j < idx.size(); ++j)
      if (mode) outs_all[idx[j]] = outs[j];
  }
  std::vector<torch::Tensor> outs_ret;
  if (mode)
    for (auto &o : outs_all) outs_ret.push_back(o);
  return {counts, outs_ret};
}


// K1/K2+K4 fused: moments + HLL registers in one read.
// Returns (moments [ncols,9] f64, regs [ncols, 1<<p] i32).
std::tuple<torch::Tensor, torch::Tensor> moments_hll(std::vector<torch::Tensor> cols, int64_t p,
                                                     std::vector<double> shifts) {
  TORCH_CHECK(!cols.empty(), "no columns");
  TORCH_CHECK(shifts.empty() || shifts.size() == cols.size(), "shifts must match cols");
  auto device = cols[0].device();
  int64_t m = 1LL << p;
  auto mom = torch::zeros({(int64_t)cols.size(), 9},
                          torch::TensorOptions().dtype(torch::kFloat64).device(device));
  auto regs = torch::zeros({(int64_t)cols.size(), m},
                           torch::TensorOptions().dtype(torch::kInt32).device(device));
  for (int pass = 0; pass < 2; ++pass) {
    std::vector<int64_t> ptrs, lens, idx;
    std::vector<double> sh;
    for (size_t i = 0; i < cols.size(); ++i) {
      auto &t = cols[i];
      TORCH_CHECK(t.is_contiguous() && t.device() == device, "columns must be contiguous, same device");
      if (dtype_code(t) != pass) continue;
      ptrs.push_back((int64_t)t.data_ptr());
      lens.push_back(t.numel());
      sh.push_back(shifts.empty() ? 0.0 : shifts[i]);
      idx.push_back((int64_t)i);
    }
    if (ptrs.empty()) continue;
    int ncols = (int)ptrs.size();
    int64_t maxn = *std::max_element(lens.begin(), lens.end());
    int nchunks = pick_chunks(maxn, ncols);
    auto dptr = to_device_i64(ptrs, device);
    auto dlen = to_device_i64(lens, device);
    auto dsh = to_device_f64(sh, device);
    auto partials = torch::empty({(int64_t)ncols * nchunks, 9},
                                 torch::TensorOptions().dtype(torch::kFloat64).device(device));
    auto mom_sub = torch::empty({ncols, 9}, torch::TensorOptions().dtype(torch::kFloat64).device(device));
    auto reg_sub = torch::zeros({ncols, m}, torch::TensorOptions().dtype(torch::kInt32).device(device));
    check_hip(anovos_moments_hll((const void *const *)dptr.data_ptr<int64_t>(),
                                 dlen.data_ptr<int64_t>(), dsh.data_ptr<double>(),
                                 ncols, (int)p, nchunks, pass,
                                 partials.data_ptr<double>(), mom_sub.data_ptr<double>(),
                                 reg_sub.data_ptr<int>(), current_stream()),
              "anovos_moments_hll");
    auto didx = to_device_i64(idx, device);
    mom.index_copy_(0, didx, mom_sub);
    regs.index_copy_(0, didx, reg_sub);
  }
  return {mom, regs};
}

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("code_counts_multi", &code_counts_multi, "fused multi-column code counts + null slot (K5)");
  m.def("outlier_clamp_columns", &outlier_clamp_columns, "fused outlier count/clamp (K10/K11)");
  m.def("label_counts_multi", &label_counts_multi, "fused multi-column label-conditioned histograms (K9)");
  m.def("bucketize_label_counts", &bucketize_label_counts, "fused bucketize + label counts, no binned materialization (K6+K9)");
  m.def("moments_hll", &moments_hll, "fused moments + HLL registers (K1/K2+K4)",
        py::arg("cols"), py::arg("p"), py::arg("shifts") = std::vector<double>());
  m.def("centered_gram_bf16", &centered_gram_bf16, "bf16 MFMA centered Gram X^T X (K8)");
  m.def("bracket_histograms_grouped", &bracket_histograms_grouped, "grouped refinement histograms (K3)");
  m.def("bucketize_columns_float", &bucketize_columns_float, "bucketize to float bin labels (K6)");
  m.def("lut_apply_f32", &lut_apply_f32, "fused LUT gather -> float (K12)");
  m.def("lut_apply_i32", &lut_apply_i32, "fused LUT gather -> int32 (K12)");
  m.def("hll_registers_multi", &hll_registers_multi, "fused multi-column HLL (K4)");
  m.def("scale_columns", &scale_columns, "fused (x-a)*b scaling (K11)");
  m.def("fill_nan_columns", &fill_nan_columns, "fused NaN fill (K11)");
  m.def("fill_code_columns", &fill_code_columns, "fused categorical null fill (K11)");
  m.def("column_moments", &column_moments, "fused per-column moments (K1/K2)",
        py::arg("cols"), py::arg("shifts") = std::vector<double>());
  m.def("column_histograms", &column_histograms, "fused per-column histograms (K3/K6)");
  m.def("bracket_histograms", &bracket_histograms, "quantile-refinement histograms (K3)");
  m.def("bucketize_columns", &bucketize_columns, "branchless bucketize (K6)");
  m.def("code_counts", &code_counts, "dictionary code bincount (K5)");
  m.def("hll_registers", &hll_registers, "HyperLogLog registers (K4)");
  m.def("row_null_counts_num", &row_null_counts_num, "fused row null counts (K10)");
}
