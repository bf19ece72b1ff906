// anovos_amd HIP kernels for MI355X (gfx950, CDNA4).
//
// Kernel inventory:
//   K1/K2  column_moments      — fused per-column count/sum/sum2/sum3/sum4/
//                                min/max/zeros, fp64 accumulators, partials
//                                per (col, chunk) block + deterministic
//                                second-stage reduce (no atomics).
//   K3/K6  column_histograms   — equal-width histograms, LDS-staged bins.
//          bracket_histograms  — quantile-refinement histograms.
//          bucketize_columns   — branchless binary search over LDS cutoffs.
//   K4     hll_registers       — HyperLogLog (p=12 default), LDS-staged
//                                registers; moments_hll fuses K1/K2+K4.
//   K5     code_counts         — dictionary-code bincount, LDS-staged.
//   K10    row_null_counts     — fused row-wise NaN count across columns.
//
// Design notes (cdna_hip_programming.md): wave64; blocks of 256 threads;
// grids sized ncols x chunks >> 256 workgroups to fill 8 XCDs; fp32 loads
// vectorized as float4 (16B/lane); all cross-block merging goes through a
// partials buffer + second kernel so results are deterministic.

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>
#include <cmath>

#define THREADS 256
#define DEV_INLINE __device__ __forceinline__

// native ext vectors for the streaming loops: the nontemporal builtins
// reject HIP_vector_type; 2x unroll + nt load/store measured +5-8% over
// plain float4 on gfx950 (tools/bwbench.hip)
typedef int nat_i4 __attribute__((ext_vector_type(4)));
typedef float nat_f4 __attribute__((ext_vector_type(4)));

// ------------------------------------------------------------------
// K1/K2: fused column moments
// ------------------------------------------------------------------
// partials layout: [ncols * nchunks][8] doubles:
//   0:n 1:s1 2:s2 3:s3 4:s4 5:min 6:max 7:zeros

// 9 slots: n, s1..s4, min, max, zeros, n_frac (non-integral count — lets
// the host pick dense-bincount exact modes for integral columns for free)
#define NSTAT 9

struct MomAcc {
  double n, s1, s2, s3, s4, mn, mx, zn, nf;
};

// s1..s4 accumulate about a per-column pivot (shift): raw power sums of
// x cancel catastrophically for skew/kurt when |mean| >> stddev (the
// SURVEY hard-part on moment stability at 1e9 rows); with a pivot taken
// from the data, d = x - shift stays O(spread) and fp64 keeps ~15
// significant digits of the central moments. min/max/zeros/n_frac use
// the raw value.
DEV_INLINE void mom_add(MomAcc &a, double v, double shift) {
  if (!isnan(v)) {
    a.n += 1.0;
    double d = v - shift;
    a.s1 += d;
    double d2 = d * d;
    a.s2 += d2;
    a.s3 += d2 * d;
    a.s4 += d2 * d2;
    a.mn = fmin(a.mn, v);
    a.mx = fmax(a.mx, v);
    a.zn += (v == 0.0) ? 1.0 : 0.0;
    a.nf += (v != trunc(v)) ? 1.0 : 0.0;
  }
}

// Block-combine a MomAcc via wave shuffles + a 4-slot LDS handoff
// (the old 256xNSTAT LDS tree cost 18 KB per block — stacked on the
// fused kernel's 16 KB HLL registers it capped occupancy at 4 blocks/CU
// and the frame read ran at 2.8 TB/s instead of ~6).
DEV_INLINE void mom_wave_reduce(MomAcc &a) {
  for (int off = 32; off > 0; off >>= 1) {
    a.n += __shfl_down(a.n, off);
    a.s1 += __shfl_down(a.s1, off);
    a.s2 += __shfl_down(a.s2, off);
    a.s3 += __shfl_down(a.s3, off);
    a.s4 += __shfl_down(a.s4, off);
    a.mn = fmin(a.mn, __shfl_down(a.mn, off));
    a.mx = fmax(a.mx, __shfl_down(a.mx, off));
    a.zn += __shfl_down(a.zn, off);
    a.nf += __shfl_down(a.nf, off);
  }
}

DEV_INLINE void mom_block_reduce_write(MomAcc &a, double *dst) {
  mom_wave_reduce(a);
  __shared__ double smw[(THREADS / 64) * NSTAT];
  const int wave = threadIdx.x >> 6;
  if ((threadIdx.x & 63) == 0) {
    double *m = &smw[wave * NSTAT];
    m[0] = a.n; m[1] = a.s1; m[2] = a.s2; m[3] = a.s3; m[4] = a.s4;
    m[5] = a.mn; m[6] = a.mx; m[7] = a.zn; m[8] = a.nf;
  }
  __syncthreads();
  if (threadIdx.x == 0) {
    MomAcc r{0, 0, 0, 0, 0, DBL_MAX, -DBL_MAX, 0, 0};
    for (int wv = 0; wv < THREADS / 64; ++wv) {
      double *m = &smw[wv * NSTAT];
      r.n += m[0]; r.s1 += m[1]; r.s2 += m[2]; r.s3 += m[3]; r.s4 += m[4];
      r.mn = fmin(r.mn, m[5]); r.mx = fmax(r.mx, m[6]);
      r.zn += m[7]; r.nf += m[8];
    }
    dst[0] = r.n; dst[1] = r.s1; dst[2] = r.s2; dst[3] = r.s3; dst[4] = r.s4;
    dst[5] = r.mn; dst[6] = r.mx; dst[7] = r.zn; dst[8] = r.nf;
  }
}

template <typename T>
__global__ __launch_bounds__(THREADS) void moments_partials_kernel(
    const T *const *cols, const int64_t *lens, const double *shifts, int nchunks,
    double *partials) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const double shift = shifts ? shifts[col] : 0.0;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);

  MomAcc a{0, 0, 0, 0, 0, DBL_MAX, -DBL_MAX, 0};

  if (sizeof(T) == 4) {
    // vectorized float4 path: 16 B per lane per iteration
    const int64_t nv = (e - s) / 4;
    const float4 *xv = reinterpret_cast<const float4 *>(x + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      float4 v = xv[i];
      mom_add(a, (double)v.x, shift);
      mom_add(a, (double)v.y, shift);
      mom_add(a, (double)v.z, shift);
      mom_add(a, (double)v.w, shift);
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) mom_add(a, (double)x[i], shift);
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) mom_add(a, (double)x[i], shift);
  }

  mom_block_reduce_write(a, &partials[(int64_t)blockIdx.x * NSTAT]);
}

__global__ __launch_bounds__(THREADS) void moments_reduce_kernel(
    const double *partials, int nchunks, double *out) {
  const int col = blockIdx.x;
  MomAcc a{0, 0, 0, 0, 0, DBL_MAX, -DBL_MAX, 0, 0};
  for (int c = threadIdx.x; c < nchunks; c += THREADS) {
    const double *p = &partials[((int64_t)col * nchunks + c) * NSTAT];
    a.n += p[0]; a.s1 += p[1]; a.s2 += p[2]; a.s3 += p[3]; a.s4 += p[4];
    a.mn = fmin(a.mn, p[5]); a.mx = fmax(a.mx, p[6]); a.zn += p[7]; a.nf += p[8];
  }
  __shared__ double sm[THREADS * NSTAT];
  double *mine = &sm[threadIdx.x * NSTAT];
  mine[0] = a.n; mine[1] = a.s1; mine[2] = a.s2; mine[3] = a.s3;
  mine[4] = a.s4; mine[5] = a.mn; mine[6] = a.mx; mine[7] = a.zn; mine[8] = a.nf;
  __syncthreads();
  for (int stride = THREADS / 2; stride > 0; stride >>= 1) {
    if (threadIdx.x < stride) {
      double *other = &sm[(threadIdx.x + stride) * NSTAT];
      mine[0] += other[0]; mine[1] += other[1]; mine[2] += other[2];
      mine[3] += other[3]; mine[4] += other[4];
      mine[5] = fmin(mine[5], other[5]);
      mine[6] = fmax(mine[6], other[6]);
      mine[7] += other[7]; mine[8] += other[8];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    double *o = &out[(int64_t)col * NSTAT];
    o[0] = sm[0]; o[1] = sm[1]; o[2] = sm[2]; o[3] = sm[3]; o[4] = sm[4];
    o[5] = (sm[0] > 0) ? sm[5] : nan("");
    o[6] = (sm[0] > 0) ? sm[6] : nan("");
    o[7] = sm[7]; o[8] = sm[8];
  }
}

// ------------------------------------------------------------------
// K3/K6: histograms (LDS bins, atomic flush to global)
// ------------------------------------------------------------------

template <typename T>
__global__ __launch_bounds__(THREADS) void hist_kernel(
    const T *const *cols, const int64_t *lens, const double *lo, const double *hi,
    int nbins, int nchunks, uint64_t *out /*[ncols][nbins]*/) {
  extern __shared__ uint32_t bins[];  // nbins
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  for (int b = threadIdx.x; b < nbins; b += THREADS) bins[b] = 0;
  __syncthreads();

  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const double l = lo[col];
  const double h = hi[col];
  if (h > l) {
    const double scale = (double)nbins / (h - l);
    auto body = [&](double v) {
      if (!isnan(v)) {
        int b = (int)((v - l) * scale);
        b = max(0, min(nbins - 1, b));
        atomicAdd(&bins[b], 1u);
      }
    };
    if (sizeof(T) == 4) {
      const int64_t nv = (e - s) / 4;
      const float4 *xv = reinterpret_cast<const float4 *>(x + s);
      for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
        float4 v = xv[i];
        body((double)v.x);
        body((double)v.y);
        body((double)v.z);
        body((double)v.w);
      }
      for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) body((double)x[i]);
    } else {
      for (int64_t i = s + threadIdx.x; i < e; i += THREADS) body((double)x[i]);
    }
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      double v = (double)x[i];
      if (!isnan(v)) atomicAdd(&bins[0], 1u);
    }
  }
  __syncthreads();
  uint64_t *g = &out[(int64_t)col * nbins];
  for (int b = threadIdx.x; b < nbins; b += THREADS)
    if (bins[b]) atomicAdd((unsigned long long *)&g[b], (unsigned long long)bins[b]);
}

// bracket histograms: like hist but per-(bracket) with col indirection and
// values outside [lo, hi) skipped.
template <typename T>
__global__ __launch_bounds__(THREADS) void bracket_hist_kernel(
    const T *const *cols, const int64_t *lens, const int64_t *colidx,
    const double *lo, const double *hi, int nbins, int nchunks, uint64_t *out) {
  extern __shared__ uint32_t bins[];
  const int bracket = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  for (int b = threadIdx.x; b < nbins; b += THREADS) bins[b] = 0;
  __syncthreads();
  const int col = (int)colidx[bracket];
  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const double l = lo[bracket];
  const double h = hi[bracket];
  if (h > l) {
    const double scale = (double)nbins / (h - l);
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      double v = (double)x[i];
      if (!isnan(v) && v >= l && v < h) {
        int b = (int)((v - l) * scale);
        b = max(0, min(nbins - 1, b));
        atomicAdd(&bins[b], 1u);
      }
    }
  }
  __syncthreads();
  uint64_t *g = &out[(int64_t)bracket * nbins];
  for (int b = threadIdx.x; b < nbins; b += THREADS)
    if (bins[b]) atomicAdd((unsigned long long *)&g[b], (unsigned long long)bins[b]);
}

// grouped bracket histograms: one column read serves ALL of that
// column's refinement brackets. Brackets live on the pass-1 bin grid, so
// an LDS LUT (pass-1 bin -> local bracket id) resolves the bracket in
// O(1) per element instead of looping the bracket list; loads are
// float4-vectorized. brackets are pre-sorted by column: col c owns
// [bstart[c], bstart[c+1]).
#define RB_BINS 512
#define RB_MAXB 16

template <typename T>
__global__ __launch_bounds__(THREADS) void bracket_hist_grouped_kernel(
    const T *const *cols, const int64_t *lens, const int *bstart, int ncols,
    const double *lo, const double *hi, const double *p1lo,
    const double *p1scale, int p1bins, int nchunks, uint64_t *out) {
  __shared__ uint32_t bins[RB_MAXB * RB_BINS];
  extern __shared__ int16_t lut[];  // [p1bins] pass-1 bin -> bracket id
  __shared__ double slo[RB_MAXB], sscale[RB_MAXB], shi[RB_MAXB];
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int b0 = bstart[col];
  const int nb = bstart[col + 1] - b0;
  if (nb == 0) return;
  const double l1 = p1lo[col];
  const double sc1 = p1scale[col];
  for (int i = threadIdx.x; i < nb * RB_BINS; i += THREADS) bins[i] = 0;
  for (int i = threadIdx.x; i < p1bins; i += THREADS) lut[i] = -1;
  __syncthreads();
  if (threadIdx.x < nb) {
    const double bl = lo[b0 + threadIdx.x];
    const double bh = hi[b0 + threadIdx.x];
    slo[threadIdx.x] = bl;
    shi[threadIdx.x] = bh;
    double w = bh - bl;
    sscale[threadIdx.x] = (w > 0) ? (double)RB_BINS / w : 0.0;
    // floor with epsilon: grid-aligned bl maps exactly; a sub-bin
    // bracket (refine >= 2) maps to the bin that CONTAINS it
    int bin1 = (int)((bl - l1) * sc1 + 1e-6);
    if (bin1 >= 0 && bin1 < p1bins) lut[bin1] = (int16_t)threadIdx.x;
  }
  __syncthreads();
  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);

  auto body = [&](double v) {
    if (isnan(v)) return;
    int b1 = (int)((v - l1) * sc1);
    b1 = max(0, min(p1bins - 1, b1));
    int br = lut[b1];
    if (br >= 0 && v >= slo[br] && v < shi[br]) {
      int bin = (int)((v - slo[br]) * sscale[br]);
      bin = max(0, min(RB_BINS - 1, bin));
      atomicAdd(&bins[br * RB_BINS + bin], 1u);
    }
  };
  if (sizeof(T) == 4) {
    const int64_t nv = (e - s) / 4;
    const float4 *xv = reinterpret_cast<const float4 *>(x + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      float4 v = xv[i];
      body((double)v.x);
      body((double)v.y);
      body((double)v.z);
      body((double)v.w);
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) body((double)x[i]);
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) body((double)x[i]);
  }
  __syncthreads();
  for (int i = threadIdx.x; i < nb * RB_BINS; i += THREADS)
    if (bins[i])
      atomicAdd((unsigned long long *)&out[(int64_t)(b0 + i / RB_BINS) * RB_BINS + (i % RB_BINS)],
                (unsigned long long)bins[i]);
}

// ------------------------------------------------------------------
// bucketize: branchless binary search over per-column cutoffs in LDS
// ------------------------------------------------------------------

template <typename T>
__global__ __launch_bounds__(THREADS) void bucketize_kernel(
    const T *const *cols, const int64_t *lens,
    const double *cutflat, const int64_t *cutoff_off, const int *cutoff_len,
    int nchunks, int32_t *const *outs) {
  extern __shared__ double cuts[];
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int ncut = cutoff_len[col];
  const double *src = &cutflat[cutoff_off[col]];
  for (int i = threadIdx.x; i < ncut; i += THREADS) cuts[i] = src[i];
  __syncthreads();

  const T *__restrict__ x = cols[col];
  int32_t *__restrict__ out = outs[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
    double v = (double)x[i];
    if (isnan(v)) {
      out[i] = -1;
      continue;
    }
    // first index with cuts[idx] >= v  (torch.bucketize right=False)
    int lo = 0, len = ncut;
    while (len > 0) {
      int half = len >> 1;
      int mid = lo + half;
      // branchless: move lo past mid when cuts[mid] < v
      lo = (cuts[mid] < v) ? (mid + 1) : lo;
      len = (cuts[mid] < v) ? (len - half - 1) : half;
    }
    out[i] = lo;
  }
}

// float-output bucketize: writes (bin index + 1) as float32 with NaN for
// null — exactly the binned-column layout attribute_binning materializes
// (avoids three extra elementwise passes per column on the python side).
template <typename T>
__global__ __launch_bounds__(THREADS) void bucketize_float_kernel(
    const T *const *cols, const int64_t *lens,
    const double *cutflat, const int64_t *cutoff_off, const int *cutoff_len,
    int nchunks, float *const *outs) {
  extern __shared__ double cuts[];
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int ncut = cutoff_len[col];
  const double *src = &cutflat[cutoff_off[col]];
  // f32 columns scan f32 cuts with f64-EXACT placement: for an f32
  // value v, (cut < (double)v) <=> (cutf < v) where cutf is (float)cut
  // adjusted DOWN to the largest f32 whose double is <= ... i.e. if
  // (double)(float)cut > cut, step one ulp down. Halves the LDS reads
  // and removes the per-element f64 convert/compare chain.
  float *cutsf = reinterpret_cast<float *>(cuts);
  if (sizeof(T) == 4) {
    for (int i = threadIdx.x; i < ncut; i += THREADS) {
      double c = src[i];
      float cf = (float)c;
      if ((double)cf > c) cf = nextafterf(cf, -(float)INFINITY);
      cutsf[i] = cf;
    }
  } else {
    for (int i = threadIdx.x; i < ncut; i += THREADS) cuts[i] = src[i];
  }
  __syncthreads();
  const T *__restrict__ x = cols[col];
  float *__restrict__ out = outs[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);

  auto place = [&](double v) -> float {
    if (isnan(v)) return nanf("");
    int lo;
    if (ncut <= 32) {
      // linear scan with a wave-uniform index: cuts[j] is an LDS
      // broadcast (no bank conflicts), the adds are branchless — beats
      // the divergent-index binary search for the typical 9-cut case
      lo = 0;
      for (int j = 0; j < ncut; ++j) lo += (cuts[j] < v) ? 1 : 0;
    } else {
      int len = ncut;
      lo = 0;
      while (len > 0) {
        int half = len >> 1;
        int mid = lo + half;
        lo = (cuts[mid] < v) ? (mid + 1) : lo;
        len = (cuts[mid] < v) ? (len - half - 1) : half;
      }
    }
    return (float)(lo + 1);
  };
  auto placef = [&](float v) -> float {
    if (isnan(v)) return nanf("");
    int lo;
    if (ncut <= 32) {
      lo = 0;
      for (int j = 0; j < ncut; ++j) lo += (cutsf[j] < v) ? 1 : 0;
    } else {
      int len = ncut;
      lo = 0;
      while (len > 0) {
        int half = len >> 1;
        int mid = lo + half;
        lo = (cutsf[mid] < v) ? (mid + 1) : lo;
        len = (cutsf[mid] < v) ? (len - half - 1) : half;
      }
    }
    return (float)(lo + 1);
  };
  if (sizeof(T) == 4) {
    const int64_t nv = (e - s) / 4;
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>(x + s);
    nat_f4 *__restrict__ ov = reinterpret_cast<nat_f4 *>(out + s);
    int64_t i = threadIdx.x;
    for (; i + THREADS < nv; i += 2 * THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      nat_f4 w = __builtin_nontemporal_load(&xv[i + THREADS]);
      nat_f4 r, q;
      r.x = placef(v.x);
      r.y = placef(v.y);
      r.z = placef(v.z);
      r.w = placef(v.w);
      q.x = placef(w.x);
      q.y = placef(w.y);
      q.z = placef(w.z);
      q.w = placef(w.w);
      __builtin_nontemporal_store(r, &ov[i]);
      __builtin_nontemporal_store(q, &ov[i + THREADS]);
    }
    for (; i < nv; i += THREADS) {
      nat_f4 v = xv[i];
      nat_f4 r;
      r.x = placef(v.x);
      r.y = placef(v.y);
      r.z = placef(v.z);
      r.w = placef(v.w);
      ov[i] = r;
    }
    for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS)
      out[j] = placef((float)x[j]);
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS)
      out[i] = place((double)x[i]);
  }
}

// ------------------------------------------------------------------
// K12: fused dictionary LUT apply (encodings / category remaps)
// ------------------------------------------------------------------
// f32 variant: out = lut[code] (float), null code (-1) -> NaN.
__global__ __launch_bounds__(THREADS) void lut_apply_f32_kernel(
    const int32_t *const *cols, const int64_t *lens, const float *lutflat,
    const int64_t *lut_off, int nchunks, float *const *outs) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int32_t *__restrict__ codes = cols[col];
  float *__restrict__ out = outs[col];
  const float *__restrict__ lut = &lutflat[lut_off[col]];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const nat_i4 *__restrict__ cv = reinterpret_cast<const nat_i4 *>(codes + s);
  nat_f4 *__restrict__ ov = reinterpret_cast<nat_f4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
    nat_i4 c = __builtin_nontemporal_load(&cv[i]);
    nat_f4 r;
    r.x = (c.x < 0) ? nanf("") : lut[c.x];
    r.y = (c.y < 0) ? nanf("") : lut[c.y];
    r.z = (c.z < 0) ? nanf("") : lut[c.z];
    r.w = (c.w < 0) ? nanf("") : lut[c.w];
    __builtin_nontemporal_store(r, &ov[i]);
  }
  for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS) {
    int c = codes[j];
    out[j] = (c < 0) ? nanf("") : lut[c];
  }
}

// i32 variant: out = lut[code] (int32 code remap), null (-1) -> -1.
__global__ __launch_bounds__(THREADS) void lut_apply_i32_kernel(
    const int32_t *const *cols, const int64_t *lens, const int32_t *lutflat,
    const int64_t *lut_off, int nchunks, int32_t *const *outs) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int32_t *__restrict__ codes = cols[col];
  int32_t *__restrict__ out = outs[col];
  const int32_t *__restrict__ lut = &lutflat[lut_off[col]];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const nat_i4 *__restrict__ cv = reinterpret_cast<const nat_i4 *>(codes + s);
  nat_i4 *__restrict__ ov = reinterpret_cast<nat_i4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
    nat_i4 c = __builtin_nontemporal_load(&cv[i]);
    nat_i4 r;
    r.x = (c.x < 0) ? -1 : lut[c.x];
    r.y = (c.y < 0) ? -1 : lut[c.y];
    r.z = (c.z < 0) ? -1 : lut[c.z];
    r.w = (c.w < 0) ? -1 : lut[c.w];
    __builtin_nontemporal_store(r, &ov[i]);
  }
  for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS) {
    int c = codes[j];
    out[j] = (c < 0) ? -1 : lut[c];
  }
}

// Fused categorical null-fill (mode imputation): out = (code == -1) ?
// fill[col] : code, one launch for every column. int4 vectorized with
// non-temporal access (streamed once).
__global__ __launch_bounds__(THREADS) void fill_code_kernel(
    const int32_t *const *cols, const int64_t *lens, const int32_t *fill,
    int nchunks, int32_t *const *outs) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int32_t *__restrict__ x = cols[col];
  int32_t *__restrict__ out = outs[col];
  const int64_t n = lens[col];
  const int32_t fv = fill[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const nat_i4 *__restrict__ xv = reinterpret_cast<const nat_i4 *>(x + s);
  nat_i4 *__restrict__ ov = reinterpret_cast<nat_i4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  int64_t i = threadIdx.x;
  for (; i + THREADS < nv; i += 2 * THREADS) {
    nat_i4 a = __builtin_nontemporal_load(&xv[i]);
    nat_i4 b = __builtin_nontemporal_load(&xv[i + THREADS]);
    a.x = (a.x == -1) ? fv : a.x;
    a.y = (a.y == -1) ? fv : a.y;
    a.z = (a.z == -1) ? fv : a.z;
    a.w = (a.w == -1) ? fv : a.w;
    b.x = (b.x == -1) ? fv : b.x;
    b.y = (b.y == -1) ? fv : b.y;
    b.z = (b.z == -1) ? fv : b.z;
    b.w = (b.w == -1) ? fv : b.w;
    __builtin_nontemporal_store(a, &ov[i]);
    __builtin_nontemporal_store(b, &ov[i + THREADS]);
  }
  for (; i < nv; i += THREADS) {
    nat_i4 a = xv[i];
    a.x = (a.x == -1) ? fv : a.x;
    a.y = (a.y == -1) ? fv : a.y;
    a.z = (a.z == -1) ? fv : a.z;
    a.w = (a.w == -1) ? fv : a.w;
    ov[i] = a;
  }
  for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS) {
    int32_t c = x[j];
    out[j] = (c == -1) ? fv : c;
  }
}

// ------------------------------------------------------------------
// K5: dictionary code bincount
// ------------------------------------------------------------------

__global__ __launch_bounds__(THREADS) void code_counts_kernel(
    const int32_t *codes, int64_t n, int size, int nchunks, uint64_t *out) {
  extern __shared__ uint32_t cnt[];
  const bool use_lds = (size <= 16384);
  if (use_lds) {
    for (int i = threadIdx.x; i < size; i += THREADS) cnt[i] = 0;
    __syncthreads();
  }
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  if (use_lds) {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      int c = codes[i];
      if (c >= 0 && c < size) atomicAdd(&cnt[c], 1u);
    }
    __syncthreads();
    for (int i = threadIdx.x; i < size; i += THREADS)
      if (cnt[i]) atomicAdd((unsigned long long *)&out[i], (unsigned long long)cnt[i]);
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      int c = codes[i];
      if (c >= 0 && c < size) atomicAdd((unsigned long long *)&out[c], 1ull);
    }
  }
}

// ------------------------------------------------------------------
// K4: HyperLogLog registers (p fixed by host, registers int32)
// ------------------------------------------------------------------

DEV_INLINE uint64_t splitmix64(uint64_t x) {
  x ^= x >> 30; x *= 0xBF58476D1CE4E5B9ull;
  x ^= x >> 27; x *= 0x94D049BB133111EBull;
  x ^= x >> 31;
  return x;
}

// hash one element: float32 hashes its raw 32 bits through the murmur3
// finalizer — int32 ALU runs at full VALU rate where the 64-bit mix is
// multi-op per step (the fused moments+HLL kernel was ALU-bound at
// 2.9 TB/s). 20 rho bits cap register values at 21 (p=12): saturation
// needs 2^20 items in one register — far beyond the 30k/register at
// 125M rows — and 32-bit value collisions stay ~1.4% at 1e8 distinct,
// inside the reference's rsd=0.05 contract. float64 keeps the 64-bit
// splitmix path (its domain exceeds 2^32).
DEV_INLINE uint32_t fmix32(uint32_t x) {
  x ^= x >> 16;
  x *= 0x85EBCA6Bu;
  x ^= x >> 13;
  x *= 0xC2B2AE35u;
  x ^= x >> 16;
  return x;
}

template <typename T>
DEV_INLINE bool hll_hash(T v, int p, int &idx, int &rho) {
  if (sizeof(T) == 4) {
    float f = (float)v;
    if (isnan(f)) return false;
    uint32_t b32;
    memcpy(&b32, &f, 4);
    uint32_t h = fmix32(b32);
    idx = (int)(h >> (32 - p));
    uint32_t rem = h << p;
    rho = (rem == 0) ? (32 - p + 1) : (__clz((int)rem) + 1);
    if (rho > 32 - p + 1) rho = 32 - p + 1;
    return true;
  }
  double d = (double)v;
  if (isnan(d)) return false;
  uint64_t bits;
  memcpy(&bits, &d, 8);
  uint64_t h = splitmix64(bits);
  idx = (int)(h >> (64 - p));
  uint64_t rem = h << p;
  rho = (rem == 0) ? (64 - p + 1) : (__clzll((long long)rem) + 1);
  if (rho > 64 - p + 1) rho = 64 - p + 1;
  return true;
}

template <typename T>
__global__ __launch_bounds__(THREADS) void hll_kernel(
    const T *x, int64_t n, int p, int nchunks, int32_t *regs /*[1<<p]*/) {
  extern __shared__ int32_t sreg[];  // 1<<p
  const int m = 1 << p;
  for (int i = threadIdx.x; i < m; i += THREADS) sreg[i] = 0;
  __syncthreads();
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  if (sizeof(T) == 4) {
    const float4 *xv = reinterpret_cast<const float4 *>(x + s);
    const int64_t nv = (e - s) / 4;
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      float4 v = xv[i];
      int idx, rho;
      if (hll_hash((T)v.x, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
      if (hll_hash((T)v.y, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
      if (hll_hash((T)v.z, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
      if (hll_hash((T)v.w, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) {
      int idx, rho;
      if (hll_hash(x[i], p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
    }
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      int idx, rho;
      if (hll_hash(x[i], p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < m; i += THREADS)
    if (sreg[i]) atomicMax(&regs[i], sreg[i]);
}

// ------------------------------------------------------------------
// K10: fused row-wise NaN count over numeric columns
// ------------------------------------------------------------------
// Grid: (row_chunk, col_chunk). Each block owns ROWS_PB rows x <=COLS_PB
// columns; in the vectorized path each thread owns fixed row-quads and
// counts in VGPRs (no LDS traffic in the loop), loads are float4 nt, and
// the block adds its partial counts to global with one atomic per row
// only when col chunking splits a row. LDS uint16 counters remain for
// the unaligned-tail fallback.

#define ROWS_PB 8192
#define COLS_PB 64

// null predicate: NaN for float/double, code -1 for int32 dictionaries
DEV_INLINE bool is_null_elem(float v) { return isnan(v); }
DEV_INLINE bool is_null_elem(double v) { return isnan(v); }
DEV_INLINE bool is_null_elem(int32_t v) { return v == -1; }

template <typename T>
__global__ __launch_bounds__(THREADS) void row_null_kernel(
    const T *const *cols, int ncols, int64_t n, int32_t *out) {
  __shared__ uint16_t rc[ROWS_PB];
  const int ncolchunks = (ncols + COLS_PB - 1) / COLS_PB;
  const int rowchunk = blockIdx.x / ncolchunks;
  const int colchunk = blockIdx.x % ncolchunks;
  const int64_t s = (int64_t)rowchunk * ROWS_PB;
  const int64_t e = min(n, s + ROWS_PB);
  const int nr = (int)(e - s);
  const int c0 = colchunk * COLS_PB;
  const int c1 = min(ncols, c0 + COLS_PB);
  const bool vec_ok = (sizeof(T) == 4) && (nr % 4 == 0) && ((s & 3) == 0);
  if (vec_ok) {
    // register accumulation: each thread owns QPT fixed row-quads and
    // keeps their counters in VGPRs — the LDS read-modify-write per
    // float4 bounded the old loop at ~3.9 TB/s
    constexpr int QPT = ROWS_PB / 4 / THREADS;  // quads per thread
    ushort4 acc[QPT];
#pragma unroll
    for (int q = 0; q < QPT; ++q) acc[q] = ushort4{0, 0, 0, 0};
    const int nq = nr / 4;
    typedef T T4 __attribute__((ext_vector_type(4)));
    for (int c = c0; c < c1; ++c) {
      const T4 *__restrict__ xv = reinterpret_cast<const T4 *>(cols[c] + s);
#pragma unroll
      for (int q = 0; q < QPT; ++q) {
        const int i = threadIdx.x + q * THREADS;
        if (i < nq) {
          T4 v = __builtin_nontemporal_load(&xv[i]);
          acc[q].x += is_null_elem((T)v.x);
          acc[q].y += is_null_elem((T)v.y);
          acc[q].z += is_null_elem((T)v.z);
          acc[q].w += is_null_elem((T)v.w);
        }
      }
    }
#pragma unroll
    for (int q = 0; q < QPT; ++q) {
      const int i = threadIdx.x + q * THREADS;
      if (i < nq) {
        const int64_t r = s + 4LL * i;
        if (ncolchunks == 1) {
          // out accumulates across per-dtype launches: read-modify-write
          out[r] += (int32_t)acc[q].x;
          out[r + 1] += (int32_t)acc[q].y;
          out[r + 2] += (int32_t)acc[q].z;
          out[r + 3] += (int32_t)acc[q].w;
        } else {
          if (acc[q].x) atomicAdd(&out[r], (int32_t)acc[q].x);
          if (acc[q].y) atomicAdd(&out[r + 1], (int32_t)acc[q].y);
          if (acc[q].z) atomicAdd(&out[r + 2], (int32_t)acc[q].z);
          if (acc[q].w) atomicAdd(&out[r + 3], (int32_t)acc[q].w);
        }
      }
    }
    return;
  }
  for (int i = threadIdx.x; i < ROWS_PB / 4; i += THREADS)
    ((uint64_t *)rc)[i] = 0;  // 4 counters per store
  __syncthreads();
  for (int c = c0; c < c1; ++c) {
    const T *__restrict__ x = cols[c] + s;
    for (int i = threadIdx.x; i < nr; i += THREADS)
      if (is_null_elem(x[i])) rc[i] += 1;
    // no per-column barrier: the i -> thread mapping is column-invariant
  }
  __syncthreads();
  if (ncolchunks == 1) {
    for (int i = threadIdx.x; i < nr; i += THREADS) out[s + i] += (int32_t)rc[i];
  } else {
    for (int i = threadIdx.x; i < nr; i += THREADS)
      if (rc[i]) atomicAdd(&out[s + i], (int32_t)rc[i]);
  }
}

// ------------------------------------------------------------------
// K4 fused: multi-column HyperLogLog
// ------------------------------------------------------------------

template <typename T>
__global__ __launch_bounds__(THREADS) void hll_multi_kernel(
    const T *const *cols, const int64_t *lens, int p, int nchunks,
    int32_t *regs /*[ncols][1<<p]*/) {
  extern __shared__ int32_t sreg[];
  const int m = 1 << p;
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  for (int i = threadIdx.x; i < m; i += THREADS) sreg[i] = 0;
  __syncthreads();
  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  if (sizeof(T) == 4) {
    const float4 *xv = reinterpret_cast<const float4 *>(x + s);
    const int64_t nv = (e - s) / 4;
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      float4 v = xv[i];
      int idx, rho;
      if (hll_hash((T)v.x, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
      if (hll_hash((T)v.y, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
      if (hll_hash((T)v.z, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
      if (hll_hash((T)v.w, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) {
      int idx, rho;
      if (hll_hash(x[i], p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
    }
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      int idx, rho;
      if (hll_hash(x[i], p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
    }
  }
  __syncthreads();
  int32_t *g = &regs[(int64_t)col * m];
  for (int i = threadIdx.x; i < m; i += THREADS)
    if (sreg[i]) atomicMax(&g[i], sreg[i]);
}

// ------------------------------------------------------------------
// K11 fused: multi-column scale/shift  out = (x - a) * b   (NaN passes)
//            and multi-column NaN fill out = isnan(x) ? fill : x
// ------------------------------------------------------------------

template <typename T>
__global__ __launch_bounds__(THREADS) void axpb_kernel(
    const T *const *cols, const int64_t *lens, const double *a, const double *b,
    int nchunks, float *const *outs) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const T *__restrict__ x = cols[col];
  float *__restrict__ out = outs[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const float av = (float)a[col];
  const float bv = (float)b[col];
  if (sizeof(T) == 4) {
    // dwordx4 loads need only 4-byte alignment on CDNA — no head guard;
    // 2x unroll + non-temporal keeps two loads in flight per lane
    // (tools/bwbench.hip: +5-8% over single plain float4)
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>(x + s);
    nat_f4 *__restrict__ ov = reinterpret_cast<nat_f4 *>(out + s);
    const int64_t nv = (e - s) / 4;
    int64_t i = threadIdx.x;
    for (; i + THREADS < nv; i += 2 * THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      nat_f4 w = __builtin_nontemporal_load(&xv[i + THREADS]);
      v = (v - av) * bv;
      w = (w - av) * bv;
      __builtin_nontemporal_store(v, &ov[i]);
      __builtin_nontemporal_store(w, &ov[i + THREADS]);
    }
    for (; i < nv; i += THREADS) {
      nat_f4 v = xv[i];
      ov[i] = (v - av) * bv;
    }
    for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS)
      out[j] = ((float)x[j] - av) * bv;
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS)
      out[i] = ((float)x[i] - av) * bv;
  }
}

template <typename T>
__global__ __launch_bounds__(THREADS) void fillnan_kernel(
    const T *const *cols, const int64_t *lens, const double *fill, int nchunks,
    T *const *outs) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const T *__restrict__ x = cols[col];
  T *__restrict__ out = outs[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const T fv = (T)fill[col];
  if (sizeof(T) == 4) {
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>(x + s);
    nat_f4 *__restrict__ ov = reinterpret_cast<nat_f4 *>(out + s);
    const float ff = (float)fv;
    const int64_t nv = (e - s) / 4;
    int64_t i = threadIdx.x;
    for (; i + THREADS < nv; i += 2 * THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      nat_f4 w = __builtin_nontemporal_load(&xv[i + THREADS]);
      v.x = isnan(v.x) ? ff : v.x;
      v.y = isnan(v.y) ? ff : v.y;
      v.z = isnan(v.z) ? ff : v.z;
      v.w = isnan(v.w) ? ff : v.w;
      w.x = isnan(w.x) ? ff : w.x;
      w.y = isnan(w.y) ? ff : w.y;
      w.z = isnan(w.z) ? ff : w.z;
      w.w = isnan(w.w) ? ff : w.w;
      __builtin_nontemporal_store(v, &ov[i]);
      __builtin_nontemporal_store(w, &ov[i + THREADS]);
    }
    for (; i < nv; i += THREADS) {
      nat_f4 v = xv[i];
      v.x = isnan(v.x) ? ff : v.x;
      v.y = isnan(v.y) ? ff : v.y;
      v.z = isnan(v.z) ? ff : v.z;
      v.w = isnan(v.w) ? ff : v.w;
      ov[i] = v;
    }
    for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS) {
      T v = x[j];
      out[j] = isnan((double)v) ? fv : v;
    }
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      T v = x[i];
      out[i] = isnan((double)v) ? fv : v;
    }
  }
}

// ------------------------------------------------------------------
// C API (host launchers). Streams come from the caller (torch stream).
// ------------------------------------------------------------------

#define LAUNCH_OK 0

extern "C" {

int anovos_moments(const void *const *cols, const int64_t *lens,
                   const double *shifts, int ncols, int nchunks,
                   int dtype /*0=f32 1=f64*/, double *partials, double *out,
                   hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  if (dtype == 0)
    hipLaunchKernelGGL(moments_partials_kernel<float>, grid, dim3(THREADS), 0, stream,
                       (const float *const *)cols, lens, shifts, nchunks, partials);
  else
    hipLaunchKernelGGL(moments_partials_kernel<double>, grid, dim3(THREADS), 0, stream,
                       (const double *const *)cols, lens, shifts, nchunks, partials);
  hipLaunchKernelGGL(moments_reduce_kernel, dim3(ncols), dim3(THREADS), 0, stream,
                     partials, nchunks, out);
  return (int)hipGetLastError();
}

int anovos_hist(const void *const *cols, const int64_t *lens, int ncols,
                const double *lo, const double *hi, int nbins, int nchunks,
                int dtype, uint64_t *out, hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  size_t lds = (size_t)nbins * 4;
  if (dtype == 0)
    hipLaunchKernelGGL(hist_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, lo, hi, nbins, nchunks, out);
  else
    hipLaunchKernelGGL(hist_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, lo, hi, nbins, nchunks, out);
  return (int)hipGetLastError();
}

int anovos_bracket_hist(const void *const *cols, const int64_t *lens,
                        const int64_t *colidx, int nbrackets, const double *lo,
                        const double *hi, int nbins, int nchunks, int dtype,
                        uint64_t *out, hipStream_t stream) {
  dim3 grid(nbrackets * nchunks);
  size_t lds = (size_t)nbins * 4;
  if (dtype == 0)
    hipLaunchKernelGGL(bracket_hist_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, colidx, lo, hi, nbins, nchunks, out);
  else
    hipLaunchKernelGGL(bracket_hist_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, colidx, lo, hi, nbins, nchunks, out);
  return (int)hipGetLastError();
}

int anovos_bucketize(const void *const *cols, const int64_t *lens, int ncols,
                     const double *cutflat, const int64_t *cutoff_off,
                     const int *cutoff_len, int max_ncut, int nchunks, int dtype,
                     int32_t *const *outs, hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  size_t lds = (size_t)max_ncut * 8;
  if (lds < 8) lds = 8;
  if (dtype == 0)
    hipLaunchKernelGGL(bucketize_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, cutflat, cutoff_off, cutoff_len, nchunks, outs);
  else
    hipLaunchKernelGGL(bucketize_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, cutflat, cutoff_off, cutoff_len, nchunks, outs);
  return (int)hipGetLastError();
}

int anovos_bracket_hist_grouped(const void *const *cols, const int64_t *lens,
                                const int *bstart, int ncols, const double *lo,
                                const double *hi, const double *p1lo,
                                const double *p1scale, int p1bins, int nchunks,
                                int dtype, uint64_t *out, hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  size_t lds = (size_t)p1bins * sizeof(int16_t);  // bin1 -> bracket LUT
  if (dtype == 0)
    hipLaunchKernelGGL(bracket_hist_grouped_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, bstart, ncols, lo, hi,
                       p1lo, p1scale, p1bins, nchunks, out);
  else
    hipLaunchKernelGGL(bracket_hist_grouped_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, bstart, ncols, lo, hi,
                       p1lo, p1scale, p1bins, nchunks, out);
  return (int)hipGetLastError();
}

int anovos_bucketize_float(const void *const *cols, const int64_t *lens, int ncols,
                           const double *cutflat, const int64_t *cutoff_off,
                           const int *cutoff_len, int max_ncut, int nchunks, int dtype,
                           float *const *outs, hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  size_t lds = (size_t)max_ncut * 8;
  if (lds < 8) lds = 8;
  if (dtype == 0)
    hipLaunchKernelGGL(bucketize_float_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, cutflat, cutoff_off, cutoff_len, nchunks, outs);
  else
    hipLaunchKernelGGL(bucketize_float_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, cutflat, cutoff_off, cutoff_len, nchunks, outs);
  return (int)hipGetLastError();
}

int anovos_lut_apply_f32(const int32_t *const *cols, const int64_t *lens, int ncols,
                         const float *lutflat, const int64_t *lut_off, int nchunks,
                         float *const *outs, hipStream_t stream) {
  hipLaunchKernelGGL(lut_apply_f32_kernel, dim3(ncols * nchunks), dim3(THREADS), 0, stream,
                     cols, lens, lutflat, lut_off, nchunks, outs);
  return (int)hipGetLastError();
}

int anovos_lut_apply_i32(const int32_t *const *cols, const int64_t *lens, int ncols,
                         const int32_t *lutflat, const int64_t *lut_off, int nchunks,
                         int32_t *const *outs, hipStream_t stream) {
  hipLaunchKernelGGL(lut_apply_i32_kernel, dim3(ncols * nchunks), dim3(THREADS), 0, stream,
                     cols, lens, lutflat, lut_off, nchunks, outs);
  return (int)hipGetLastError();
}

int anovos_fill_code(const int32_t *const *cols, const int64_t *lens, int ncols,
                     const int32_t *fill, int nchunks, int32_t *const *outs,
                     hipStream_t stream) {
  hipLaunchKernelGGL(fill_code_kernel, dim3(ncols * nchunks), dim3(THREADS), 0, stream,
                     cols, lens, fill, nchunks, outs);
  return (int)hipGetLastError();
}

int anovos_code_counts(const int32_t *codes, int64_t n, int size, int nchunks,
                       uint64_t *out, hipStream_t stream) {
  size_t lds = (size <= 16384) ? (size_t)size * 4 : 0;
  hipLaunchKernelGGL(code_counts_kernel, dim3(nchunks), dim3(THREADS), lds, stream,
                     codes, n, size, nchunks, out);
  return (int)hipGetLastError();
}

int anovos_hll(const void *x, int64_t n, int p, int nchunks, int dtype,
               int32_t *regs, hipStream_t stream) {
  size_t lds = (size_t)(1 << p) * 4;
  if (dtype == 0)
    hipLaunchKernelGGL(hll_kernel<float>, dim3(nchunks), dim3(THREADS), lds, stream,
                       (const float *)x, n, p, nchunks, regs);
  else
    hipLaunchKernelGGL(hll_kernel<double>, dim3(nchunks), dim3(THREADS), lds, stream,
                       (const double *)x, n, p, nchunks, regs);
  return (int)hipGetLastError();
}

int anovos_row_null(const void *const *cols, int ncols, int64_t n, int dtype,
                    int32_t *out, hipStream_t stream) {
  if (n <= 0 || ncols <= 0) return 0;  // empty shard (ranks > part files)
  const int ncolchunks = (ncols + COLS_PB - 1) / COLS_PB;
  const int64_t nrowchunks = (n + ROWS_PB - 1) / ROWS_PB;
  dim3 grid((uint32_t)(nrowchunks * ncolchunks));
  if (dtype == 0)
    hipLaunchKernelGGL(row_null_kernel<float>, grid, dim3(THREADS), 0, stream,
                       (const float *const *)cols, ncols, n, out);
  else if (dtype == 1)
    hipLaunchKernelGGL(row_null_kernel<double>, grid, dim3(THREADS), 0, stream,
                       (const double *const *)cols, ncols, n, out);
  else
    hipLaunchKernelGGL(row_null_kernel<int32_t>, grid, dim3(THREADS), 0, stream,
                       (const int32_t *const *)cols, ncols, n, out);
  return (int)hipGetLastError();
}

int anovos_hll_multi(const void *const *cols, const int64_t *lens, int ncols,
                     int p, int nchunks, int dtype, int32_t *regs,
                     hipStream_t stream) {
  size_t lds = (size_t)(1 << p) * 4;
  dim3 grid(ncols * nchunks);
  if (dtype == 0)
    hipLaunchKernelGGL(hll_multi_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, p, nchunks, regs);
  else
    hipLaunchKernelGGL(hll_multi_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, p, nchunks, regs);
  return (int)hipGetLastError();
}

int anovos_axpb(const void *const *cols, const int64_t *lens, int ncols,
                const double *a, const double *b, int nchunks, int dtype,
                float *const *outs, hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  if (dtype == 0)
    hipLaunchKernelGGL(axpb_kernel<float>, grid, dim3(THREADS), 0, stream,
                       (const float *const *)cols, lens, a, b, nchunks, outs);
  else
    hipLaunchKernelGGL(axpb_kernel<double>, grid, dim3(THREADS), 0, stream,
                       (const double *const *)cols, lens, a, b, nchunks, outs);
  return (int)hipGetLastError();
}

int anovos_fillnan(const void *const *cols, const int64_t *lens, int ncols,
                   const double *fill, int nchunks, int dtype,
                   void *const *outs, hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  if (dtype == 0)
    hipLaunchKernelGGL(fillnan_kernel<float>, grid, dim3(THREADS), 0, stream,
                       (const float *const *)cols, lens, fill, nchunks,
                       (float *const *)outs);
  else
    hipLaunchKernelGGL(fillnan_kernel<double>, grid, dim3(THREADS), 0, stream,
                       (const double *const *)cols, lens, fill, nchunks,
                       (double *const *)outs);
  return (int)hipGetLastError();
}

}  // extern "C"

// ------------------------------------------------------------------
// K5 (fused): multi-column dictionary code counts + null counts.
// One launch for ALL categorical columns: grid = ncols x nchunks;
// per-column counts land at out[off[col] .. off[col]+size[col]] and the
// null count (code < 0) at out[off[col]+size[col]]. LDS-staged when the
// dictionary fits (includes the null slot).
// ------------------------------------------------------------------
__global__ __launch_bounds__(THREADS) void code_counts_multi_kernel(
    const int32_t *const *cols, const int64_t *lens, const int64_t *offs,
    const int *sizes, int nchunks, uint64_t *out) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int32_t *__restrict__ codes = cols[col];
  const int64_t n = lens[col];
  const int size = sizes[col];
  const int slots = size + 1;  // + null slot
  uint64_t *base = out + offs[col];

  extern __shared__ uint32_t cnt[];
  const bool use_lds = (slots <= 16384);
  if (use_lds) {
    for (int i = threadIdx.x; i < slots; i += THREADS) cnt[i] = 0;
    __syncthreads();
  }
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  if (use_lds) {
    // int4 vectorized reads (the scalar loop measured ~3.3 TB/s; dword
    // loads are 1/4 the per-instruction bytes of dwordx4)
    const int64_t nv = (e - s) / 4;
    const nat_i4 *__restrict__ cv = reinterpret_cast<const nat_i4 *>(codes + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      nat_i4 c = __builtin_nontemporal_load(&cv[i]);
      atomicAdd(&cnt[(c.x >= 0 && c.x < size) ? c.x : size], 1u);
      atomicAdd(&cnt[(c.y >= 0 && c.y < size) ? c.y : size], 1u);
      atomicAdd(&cnt[(c.z >= 0 && c.z < size) ? c.z : size], 1u);
      atomicAdd(&cnt[(c.w >= 0 && c.w < size) ? c.w : size], 1u);
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) {
      int c = codes[i];
      atomicAdd(&cnt[(c >= 0 && c < size) ? c : size], 1u);
    }
    __syncthreads();
    for (int i = threadIdx.x; i < slots; i += THREADS)
      if (cnt[i]) atomicAdd((unsigned long long *)&base[i], (unsigned long long)cnt[i]);
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      int c = codes[i];
      atomicAdd((unsigned long long *)&base[(c >= 0 && c < size) ? c : size], 1ull);
    }
  }
}

// ------------------------------------------------------------------
// K10/K11 (fused): outlier flag counts + clamp/null treatment, all
// columns in one launch. lo/hi are per-column bounds (NaN = unbounded
// side). mode 0 = count only; 1 = clamp to bound; 2 = replace with NaN.
// counts[col*2+0] = lower outliers, [col*2+1] = upper. Counts use integer
// atomics (order-independent => deterministic).
// ------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(THREADS) void outlier_clamp_kernel(
    const T *const *cols, const int64_t *lens, const double *lo,
    const double *hi, int nchunks, int mode, T *const *outs,
    uint64_t *counts) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const T *__restrict__ x = cols[col];
  T *__restrict__ y = outs ? outs[col] : nullptr;
  const int64_t n = lens[col];
  const double l = lo[col], h = hi[col];
  const bool has_l = !isnan(l), has_h = !isnan(h);
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);

  uint32_t nl = 0, nh = 0;
  // branchless: NaN compares false on both sides, so the selects are
  // safe without an explicit isnan guard; the replacement value is
  // wave-uniform (lrep/hrep picked once per block from mode)
  const double lrep = (mode == 2) ? (double)NAN : l;
  const double hrep = (mode == 2) ? (double)NAN : h;
  auto clamp1 = [&](double v) -> double {
    const bool low = has_l && (v < l);
    const bool high = has_h && (v > h);
    nl += low ? 1u : 0u;
    nh += high ? 1u : 0u;
    if (mode) v = low ? lrep : (high ? hrep : v);
    return v;
  };
  if (sizeof(T) == 4) {
    // f32 fast path with f64-EXACT semantics: for an f32 value v,
    // (double)v < l  <=>  v < lf  where lf is the smallest f32 > l (or
    // l itself when representable); likewise v > h <=> v > hf with hf
    // the largest-f32-below adjustment. This drops the per-element
    // f64 convert/compare chain that ALU-bound the kernel at ~3 TB/s.
    float lf = (float)l;
    if ((double)lf < l) lf = nextafterf(lf, (float)INFINITY);
    float hf = (float)h;
    if ((double)hf > h) hf = nextafterf(hf, -(float)INFINITY);
    const float lrepf = (mode == 2) ? nanf("") : (float)l;
    const float hrepf = (mode == 2) ? nanf("") : (float)h;
    auto clamp1f = [&](float v) -> float {
      const bool low = has_l && (v < lf);
      const bool high = has_h && (v > hf);
      nl += low ? 1u : 0u;
      nh += high ? 1u : 0u;
      if (mode) v = low ? lrepf : (high ? hrepf : v);
      return v;
    };
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>(x + s);
    nat_f4 *__restrict__ ov = y ? reinterpret_cast<nat_f4 *>((float *)y + s) : nullptr;
    const int64_t nv = (e - s) / 4;
    int64_t i = threadIdx.x;
    for (; i + THREADS < nv; i += 2 * THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      nat_f4 w = __builtin_nontemporal_load(&xv[i + THREADS]);
      v.x = clamp1f(v.x);
      v.y = clamp1f(v.y);
      v.z = clamp1f(v.z);
      v.w = clamp1f(v.w);
      w.x = clamp1f(w.x);
      w.y = clamp1f(w.y);
      w.z = clamp1f(w.z);
      w.w = clamp1f(w.w);
      if (mode && y) {
        __builtin_nontemporal_store(v, &ov[i]);
        __builtin_nontemporal_store(w, &ov[i + THREADS]);
      }
    }
    for (; i < nv; i += THREADS) {
      nat_f4 v = xv[i];
      v.x = clamp1f(v.x);
      v.y = clamp1f(v.y);
      v.z = clamp1f(v.z);
      v.w = clamp1f(v.w);
      if (mode && y) ov[i] = v;
    }
    for (int64_t j = s + nv * 4 + threadIdx.x; j < e; j += THREADS) {
      float v = clamp1f((float)x[j]);
      if (mode && y) y[j] = (T)v;
    }
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      double v = clamp1((double)x[i]);
      if (mode && y) y[i] = (T)v;
    }
  }
  __shared__ uint32_t red[2][THREADS];
  red[0][threadIdx.x] = nl;
  red[1][threadIdx.x] = nh;
  __syncthreads();
  for (int st = THREADS / 2; st > 0; st >>= 1) {
    if (threadIdx.x < st) {
      red[0][threadIdx.x] += red[0][threadIdx.x + st];
      red[1][threadIdx.x] += red[1][threadIdx.x + st];
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) {
    if (red[0][0]) atomicAdd((unsigned long long *)&counts[col * 2 + 0], (unsigned long long)red[0][0]);
    if (red[1][0]) atomicAdd((unsigned long long *)&counts[col * 2 + 1], (unsigned long long)red[1][0]);
  }
}

extern "C" {

int anovos_code_counts_multi(const int32_t *const *cols, const int64_t *lens,
                             const int64_t *offs, const int *sizes, int ncols,
                             int max_slots, int nchunks, uint64_t *out,
                             hipStream_t stream) {
  size_t lds = (max_slots <= 16384) ? (size_t)max_slots * 4 : 0;
  hipLaunchKernelGGL(code_counts_multi_kernel, dim3(ncols * nchunks),
                     dim3(THREADS), lds, stream, cols, lens, offs, sizes,
                     nchunks, out);
  return (int)hipGetLastError();
}

int anovos_outlier_clamp(const void *const *cols, const int64_t *lens,
                         int ncols, const double *lo, const double *hi,
                         int nchunks, int mode, int dtype, void *const *outs,
                         uint64_t *counts, hipStream_t stream) {
  if (dtype == 0)
    hipLaunchKernelGGL(outlier_clamp_kernel<float>, dim3(ncols * nchunks),
                       dim3(THREADS), 0, stream, (const float *const *)cols,
                       lens, lo, hi, nchunks, mode, (float *const *)outs,
                       counts);
  else
    hipLaunchKernelGGL(outlier_clamp_kernel<double>, dim3(ncols * nchunks),
                       dim3(THREADS), 0, stream, (const double *const *)cols,
                       lens, lo, hi, nchunks, mode, (double *const *)outs,
                       counts);
  return (int)hipGetLastError();
}

}  // extern "C"

// ------------------------------------------------------------------
// K1/K2+K4 fused: one read of each column produces BOTH the moment
// partials and the HyperLogLog registers (moments and distinct-count
// are the two prior-free full-frame passes of the analyzer — the
// equal-width histogram can't join: its range comes FROM the moments).
// ------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(THREADS) void moments_hll_kernel(
    const T *const *cols, const int64_t *lens, const double *shifts, int p,
    int nchunks, double *partials, int32_t *regs /*[ncols][1<<p]*/) {
  extern __shared__ int32_t sreg[];  // 1<<p
  const int m = 1 << p;
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  for (int i = threadIdx.x; i < m; i += THREADS) sreg[i] = 0;
  __syncthreads();

  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const double shift = shifts ? shifts[col] : 0.0;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);

  MomAcc a{0, 0, 0, 0, 0, DBL_MAX, -DBL_MAX, 0};
  auto body = [&](T v) {
    mom_add(a, (double)v, shift);
    int idx, rho;
    if (hll_hash(v, p, idx, rho)) { if (sreg[idx] < rho) atomicMax(&sreg[idx], rho); }
  };
  if (sizeof(T) == 4) {
    const int64_t nv = (e - s) / 4;
    const float4 *xv = reinterpret_cast<const float4 *>(x + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      float4 v = xv[i];
      body((T)v.x);
      body((T)v.y);
      body((T)v.z);
      body((T)v.w);
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) body(x[i]);
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) body(x[i]);
  }

  // moment block-reduce: wave shuffles + 4-slot LDS (keeps the block's
  // LDS at the 16 KB HLL registers -> ~8 blocks/CU occupancy)
  mom_block_reduce_write(a, &partials[(int64_t)blockIdx.x * NSTAT]);
  int32_t *g = &regs[(int64_t)col * m];
  for (int i = threadIdx.x; i < m; i += THREADS)
    if (sreg[i]) atomicMax(&g[i], sreg[i]);
}

extern "C" int anovos_moments_hll(const void *const *cols, const int64_t *lens,
                                  const double *shifts, int ncols, int p,
                                  int nchunks, int dtype, double *partials,
                                  double *mom_out, int32_t *regs,
                                  hipStream_t stream) {
  dim3 grid(ncols * nchunks);
  size_t lds = (size_t)(1 << p) * 4;
  if (dtype == 0)
    hipLaunchKernelGGL(moments_hll_kernel<float>, grid, dim3(THREADS), lds, stream,
                       (const float *const *)cols, lens, shifts, p, nchunks, partials, regs);
  else
    hipLaunchKernelGGL(moments_hll_kernel<double>, grid, dim3(THREADS), lds, stream,
                       (const double *const *)cols, lens, shifts, p, nchunks, partials, regs);
  hipLaunchKernelGGL(moments_reduce_kernel, dim3(ncols), dim3(THREADS), 0, stream,
                     partials, nchunks, mom_out);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------
// K9 (fused): label-conditioned histograms for ALL columns in one
// launch (IV/IG per-bin event counts — reference
// association_evaluator.py:368-409, :533-572). Layout per column at
// out[off[col]]: [slots] total counts then [slots] event counts
// (n0 = total - event). dtypes: 0 = f32 binned values (NaN -> slot 0,
// v -> trunc(v)+1, clamped — matches the host torch fallback exactly);
// 1 = int32 dictionary codes (null/invalid -> last slot).
// ------------------------------------------------------------------
__global__ __launch_bounds__(THREADS) void label_counts_multi_kernel(
    const void *const *cols, const uint8_t *__restrict__ label,
    const int64_t *lens, const int64_t *offs, const int *sizes,
    const int *dtypes, int nchunks, uint64_t *out) {
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int slots = sizes[col];
  uint64_t *base = out + offs[col];
  extern __shared__ uint32_t lds2[];  // [2*slots]: totals then events
  uint32_t *tot = lds2;
  uint32_t *evt = lds2 + slots;
  for (int i = threadIdx.x; i < 2 * slots; i += THREADS) lds2[i] = 0;
  __syncthreads();
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  if (dtypes[col] == 0) {
    const float *__restrict__ x = (const float *)cols[col];
    const int64_t nv = (e - s) / 4;
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>(x + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      const int64_t r = s + i * 4;
      float vv[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        int slot = (vv[k] != vv[k]) ? 0 : (int)vv[k] + 1;
        slot = (slot < 0) ? 0 : ((slot >= slots) ? slots - 1 : slot);
        atomicAdd(&tot[slot], 1u);
        if (label[r + k]) atomicAdd(&evt[slot], 1u);
      }
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) {
      float v = x[i];
      int slot = (v != v) ? 0 : (int)v + 1;
      slot = (slot < 0) ? 0 : ((slot >= slots) ? slots - 1 : slot);
      atomicAdd(&tot[slot], 1u);
      if (label[i]) atomicAdd(&evt[slot], 1u);
    }
  } else {
    const int32_t *__restrict__ codes = (const int32_t *)cols[col];
    const int64_t nv = (e - s) / 4;
    const nat_i4 *__restrict__ cv = reinterpret_cast<const nat_i4 *>(codes + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      nat_i4 c = __builtin_nontemporal_load(&cv[i]);
      const int64_t r = s + i * 4;
      int cc[4] = {c.x, c.y, c.z, c.w};
#pragma unroll
      for (int k = 0; k < 4; ++k) {
        int slot = (cc[k] >= 0 && cc[k] < slots - 1) ? cc[k] : slots - 1;
        atomicAdd(&tot[slot], 1u);
        if (label[r + k]) atomicAdd(&evt[slot], 1u);
      }
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) {
      int c = codes[i];
      int slot = (c >= 0 && c < slots - 1) ? c : slots - 1;
      atomicAdd(&tot[slot], 1u);
      if (label[i]) atomicAdd(&evt[slot], 1u);
    }
  }
  __syncthreads();
  for (int i = threadIdx.x; i < slots; i += THREADS) {
    if (tot[i]) atomicAdd((unsigned long long *)&base[i], (unsigned long long)tot[i]);
    if (evt[i]) atomicAdd((unsigned long long *)&base[slots + i], (unsigned long long)evt[i]);
  }
}

extern "C" int anovos_label_counts_multi(
    const void *const *cols, const uint8_t *label, const int64_t *lens,
    const int64_t *offs, const int *sizes, const int *dtypes, int ncols,
    int max_slots, int nchunks, uint64_t *out, hipStream_t stream) {
  size_t lds = (size_t)max_slots * 2 * sizeof(uint32_t);
  hipLaunchKernelGGL(label_counts_multi_kernel, dim3(ncols * nchunks),
                     dim3(THREADS), lds, stream, cols, label, lens, offs,
                     sizes, dtypes, nchunks, out);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------
// K6+K9 (fused): bucketize + label-conditioned counts WITHOUT
// materializing the binned column. IV/IG only consume per-bin label
// counts (reference association_evaluator.py:368-409), so binning the
// frame (75 GB write + re-read at the bench shard) is pure waste: this
// kernel reads the RAW numeric column once, places each value against
// the cutoffs (identical placement to bucketize_float_kernel, incl.
// the nextafter-adjusted f32 fast path), and accumulates (total,event)
// per slot. Slot mapping matches the python fallback over a binned
// column: NaN -> 0, bin b=lo+1 -> slot b+1. Layout per column at
// out[off]: [slots] totals then [slots] events.
// ------------------------------------------------------------------
template <typename T>
__global__ __launch_bounds__(THREADS) void bucketize_label_counts_kernel(
    const T *const *cols, const uint8_t *__restrict__ label,
    const int64_t *lens, const double *cutflat, const int64_t *cutoff_off,
    const int *cutoff_len, const int64_t *offs, const int *sizes,
    int max_ncut, int max_slots, int nchunks, uint64_t *out) {
  extern __shared__ double smem[];
  const int col = blockIdx.x / nchunks;
  const int chunk = blockIdx.x % nchunks;
  const int ncut = cutoff_len[col];
  const int slots = sizes[col];
  double *cuts = smem;  // [max_ncut] doubles (f32 path reuses as floats)
  float *cutsf = reinterpret_cast<float *>(cuts);
  uint32_t *cnt = reinterpret_cast<uint32_t *>(smem + max_ncut);
  uint32_t *tot = cnt;
  uint32_t *evt = cnt + slots;
  const double *src = &cutflat[cutoff_off[col]];
  if (sizeof(T) == 4) {
    for (int i = threadIdx.x; i < ncut; i += THREADS) {
      double c = src[i];
      float cf = (float)c;
      if ((double)cf > c) cf = nextafterf(cf, -(float)INFINITY);
      cutsf[i] = cf;
    }
  } else {
    for (int i = threadIdx.x; i < ncut; i += THREADS) cuts[i] = src[i];
  }
  for (int i = threadIdx.x; i < 2 * slots; i += THREADS) cnt[i] = 0;
  __syncthreads();

  const T *__restrict__ x = cols[col];
  const int64_t n = lens[col];
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);

  auto slot_f32 = [&](float v) -> int {
    if (v != v) return 0;
    int lo = 0;
    if (ncut <= 32) {
      for (int j = 0; j < ncut; ++j) lo += (cutsf[j] < v) ? 1 : 0;
    } else {
      int len = ncut;
      while (len > 0) {
        int half = len >> 1;
        int mid = lo + half;
        lo = (cutsf[mid] < v) ? (mid + 1) : lo;
        len = (cutsf[mid] < v) ? (len - half - 1) : half;
      }
    }
    const int sl = lo + 2;  // bin (lo+1) -> slot bin+1
    return sl < slots ? sl : slots - 1;
  };
  auto slot_f64 = [&](double v) -> int {
    if (v != v) return 0;
    int lo = 0;
    if (ncut <= 32) {
      for (int j = 0; j < ncut; ++j) lo += (cuts[j] < v) ? 1 : 0;
    } else {
      int len = ncut;
      while (len > 0) {
        int half = len >> 1;
        int mid = lo + half;
        lo = (cuts[mid] < v) ? (mid + 1) : lo;
        len = (cuts[mid] < v) ? (len - half - 1) : half;
      }
    }
    const int sl = lo + 2;
    return sl < slots ? sl : slots - 1;
  };

  if (sizeof(T) == 4 && max_slots <= 16 && (e - s) / THREADS < 65000) {
    // (condition on max_slots: the launcher sizes the private LDS area
    // for the whole launch, so the branch must be launch-uniform)
    // (the chunk-size guard keeps per-thread packed u16 counters from
    // overflowing; oversized chunks fall back to the atomic path)
    // per-THREAD private LDS counters: ~12 hot slots make shared-array
    // atomics serialize on matching addresses (measured 2.1 TB/s) and
    // both the register-histogram (32 VALU/elem) and wave-ballot
    // variants measured slower. Each thread owns a padded 17-word slot
    // array (stride 17 is coprime to the 32 banks -> conflict-free) and
    // packs (event<<16 | total) into ONE u32 read-add-write per element
    // (per-thread chunk counts stay < 2^16). One combining flush at the
    // end.
    uint32_t *priv = reinterpret_cast<uint32_t *>(smem + max_ncut) + 2 * slots +
                     (uint32_t)threadIdx.x * 17;
    {
      uint32_t *base0 = reinterpret_cast<uint32_t *>(smem + max_ncut) + 2 * slots;
      for (int i = threadIdx.x; i < THREADS * 17; i += THREADS) base0[i] = 0;
    }
    __syncthreads();
    const int64_t nv = (e - s) / 4;
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>((const float *)x + s);
    // two vectors in flight per iteration (same MLP shape as
    // bucketize_float_kernel, which sustains 4.5 TB/s)
    int64_t i = threadIdx.x;
    for (; i + THREADS < nv; i += 2 * THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      nat_f4 w = __builtin_nontemporal_load(&xv[i + THREADS]);
      const int64_t r1 = s + i * 4;
      const int64_t r2 = s + (i + THREADS) * 4;
      uint32_t l1, l2;
      __builtin_memcpy(&l1, label + r1, 4);  // 4 packed label bytes
      __builtin_memcpy(&l2, label + r2, 4);
      const float vv[4] = {v.x, v.y, v.z, v.w};
      const float ww[4] = {w.x, w.y, w.z, w.w};
#pragma unroll
      for (int k2 = 0; k2 < 4; ++k2) {
        priv[slot_f32(vv[k2])] += 1u + ((uint32_t)((l1 >> (8 * k2)) & 0xFF ? 1 : 0) << 16);
        priv[slot_f32(ww[k2])] += 1u + ((uint32_t)((l2 >> (8 * k2)) & 0xFF ? 1 : 0) << 16);
      }
    }
    for (; i < nv; i += THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      const int64_t r = s + i * 4;
      const float vv[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
      for (int k2 = 0; k2 < 4; ++k2) {
        const int slot = slot_f32(vv[k2]);
        priv[slot] += 1u + ((uint32_t)(label[r + k2] != 0) << 16);
      }
    }
    for (int64_t j2 = s + nv * 4 + threadIdx.x; j2 < e; j2 += THREADS) {
      const int slot = slot_f32((float)x[j2]);
      priv[slot] += 1u + ((uint32_t)(label[j2] != 0) << 16);
    }
    __syncthreads();
    for (int s2 = 0; s2 < slots; ++s2) {
      const uint32_t v = priv[s2];
      if (v) {
        atomicAdd(&tot[s2], v & 0xFFFFu);
        atomicAdd(&evt[s2], v >> 16);
      }
    }
  } else if (sizeof(T) == 4) {
    const int64_t nv = (e - s) / 4;
    const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>((const float *)x + s);
    for (int64_t i = threadIdx.x; i < nv; i += THREADS) {
      nat_f4 v = __builtin_nontemporal_load(&xv[i]);
      const int64_t r = s + i * 4;
      const float vv[4] = {v.x, v.y, v.z, v.w};
#pragma unroll
      for (int k2 = 0; k2 < 4; ++k2) {
        const int slot = slot_f32(vv[k2]);
        atomicAdd(&tot[slot], 1u);
        if (label[r + k2]) atomicAdd(&evt[slot], 1u);
      }
    }
    for (int64_t i = s + nv * 4 + threadIdx.x; i < e; i += THREADS) {
      const int slot = slot_f32((float)x[i]);
      atomicAdd(&tot[slot], 1u);
      if (label[i]) atomicAdd(&evt[slot], 1u);
    }
  } else {
    for (int64_t i = s + threadIdx.x; i < e; i += THREADS) {
      const int slot = slot_f64((double)x[i]);
      atomicAdd(&tot[slot], 1u);
      if (label[i]) atomicAdd(&evt[slot], 1u);
    }
  }
  __syncthreads();
  uint64_t *base = out + offs[col];
  for (int i = threadIdx.x; i < slots; i += THREADS) {
    if (tot[i]) atomicAdd((unsigned long long *)&base[i], (unsigned long long)tot[i]);
    if (evt[i]) atomicAdd((unsigned long long *)&base[slots + i], (unsigned long long)evt[i]);
  }
}

extern "C" int anovos_bucketize_label_counts(
    const void *const *cols, const uint8_t *label, const int64_t *lens,
    const double *cutflat, const int64_t *cutoff_off, const int *cutoff_len,
    const int64_t *offs, const int *sizes, int ncols, int max_ncut,
    int max_slots, int nchunks, int dtype, uint64_t *out,
    hipStream_t stream) {
  size_t lds = (size_t)max_ncut * sizeof(double) +
               (size_t)max_slots * 2 * sizeof(uint32_t);
  if (dtype == 0 && max_slots <= 16)
    lds += (size_t)THREADS * 17 * sizeof(uint32_t);  // per-thread privates
  if (dtype == 0)
    hipLaunchKernelGGL(bucketize_label_counts_kernel<float>,
                       dim3(ncols * nchunks), dim3(THREADS), lds, stream,
                       (const float *const *)cols, label, lens, cutflat,
                       cutoff_off, cutoff_len, offs, sizes, max_ncut, max_slots,
                       nchunks, out);
  else
    hipLaunchKernelGGL(bucketize_label_counts_kernel<double>,
                       dim3(ncols * nchunks), dim3(THREADS), lds, stream,
                       (const double *const *)cols, label, lens, cutflat,
                       cutoff_off, cutoff_len, offs, sizes, max_ncut, max_slots,
                       nchunks, out);
  return (int)hipGetLastError();
}
