// K8: centered Gram matrix (correlation/covariance) on MFMA matrix cores.
//
// gram[i][j] = sum_r (x_i[r] - mean_i)(x_j[r] - mean_j), NaN -> 0 contribution.
// Reference semantics: association_evaluator.py:118-123 (MLlib Correlation
// .corr = one pass over the assembled vector column); here the pass is a
// tall-skinny X^T X on v_mfma_f32_16x16x32_bf16 (bf16 in, fp32 accumulate).
//
// Shape: k columns (tens..hundreds) x n rows (millions) -> HBM-bound
// (read n*k floats); MFMA keeps compute off the critical path.
//
// Tiling: columns padded to 16-col tiles; one workgroup per
// (tile-pair (i,j<=i? no: j>=i), row-chunk). Each of the 4 waves strides
// 32-row MFMA steps across the chunk (wave w rows r0+32w, step 128) and
// accumulates its own f32x4 fragment; fragments are loaded STRAIGHT from
// global (8 consecutive rows of one column = 16 B per lane, coalesced
// across the 16 lanes of a row-group) — no LDS staging needed because
// each element is touched once per tile-pair. Per-block partials land in
// a [pairs*chunks][256] buffer; a second deterministic kernel reduces
// chunks in fp64 and mirrors the symmetric half (same no-atomics
// convention as anovos_kernels.hip).

#include <hip/hip_runtime.h>
#include <cfloat>
#include <cstdint>
#include <cmath>

#define THREADS 256

using bf16x8 = __attribute__((ext_vector_type(8))) short;
using f32x4 = __attribute__((ext_vector_type(4))) float;

__device__ __forceinline__ short to_bf16(float v) {
  union {
    float f;
    uint32_t u;
  } c{v};
  // round-to-nearest-even into the upper 16 bits
  uint32_t r = c.u + 0x7FFFu + ((c.u >> 16) & 1u);
  return (short)(r >> 16);
}

// Load one lane's A/B fragment: 8 consecutive rows of column `col`,
// centered, NaN->0, zero outside [0,k) x [0,limit).
__device__ __forceinline__ bf16x8 load_frag(const float *const *cols, int k,
                                            const float *means, int col,
                                            int64_t row, int64_t limit) {
  bf16x8 f;
  if (col < k) {
    const float *__restrict__ x = cols[col];
    const float m = means[col];
    if (row + 8 <= limit) {
      const float4 *p = reinterpret_cast<const float4 *>(x + row);
      float4 a = p[0], b = p[1];
      float v0 = a.x - m, v1 = a.y - m, v2 = a.z - m, v3 = a.w - m;
      float v4 = b.x - m, v5 = b.y - m, v6 = b.z - m, v7 = b.w - m;
      f[0] = to_bf16(isnan(v0) ? 0.f : v0);
      f[1] = to_bf16(isnan(v1) ? 0.f : v1);
      f[2] = to_bf16(isnan(v2) ? 0.f : v2);
      f[3] = to_bf16(isnan(v3) ? 0.f : v3);
      f[4] = to_bf16(isnan(v4) ? 0.f : v4);
      f[5] = to_bf16(isnan(v5) ? 0.f : v5);
      f[6] = to_bf16(isnan(v6) ? 0.f : v6);
      f[7] = to_bf16(isnan(v7) ? 0.f : v7);
    } else {
#pragma unroll
      for (int e = 0; e < 8; ++e) {
        float v = (row + e < limit) ? x[row + e] - m : 0.f;
        f[e] = to_bf16(isnan(v) ? 0.f : v);
      }
    }
  } else {
#pragma unroll
    for (int e = 0; e < 8; ++e) f[e] = 0;
  }
  return f;
}

__global__ __launch_bounds__(THREADS) void gram_partials_kernel(
    const float *const *cols, int64_t n, int k, const float *means,
    const int *pair_i, const int *pair_j, int row_chunks, float *partials) {
  const int pair = blockIdx.x / row_chunks;
  const int chunk = blockIdx.x % row_chunks;
  const int i0 = pair_i[pair] * 16;
  const int j0 = pair_j[pair] * 16;

  // 32-row-aligned chunk bounds
  const int64_t steps_total = (n + 31) / 32;
  const int64_t steps_per = (steps_total + row_chunks - 1) / row_chunks;
  const int64_t step_s = (int64_t)chunk * steps_per;
  const int64_t step_e = min(steps_total, step_s + steps_per);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int m = lane & 15;         // tile row (A) / tile col (B)
  const int ko = (lane >> 4) * 8;  // K offset within the 32-row step

  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  for (int64_t s = step_s + wave; s < step_e; s += 4) {
    const int64_t r = s * 32 + ko;
    bf16x8 a = load_frag(cols, k, means, i0 + m, r, n);
    bf16x8 b = (j0 == i0) ? a : load_frag(cols, k, means, j0 + m, r, n);
    acc = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc, 0, 0, 0);
  }

  // combine the 4 waves' [16,16] fragments via LDS (C/D layout:
  // col = lane&15, row = (lane>>4)*4 + reg)
  __shared__ float lds[4][256];
#pragma unroll
  for (int reg = 0; reg < 4; ++reg) {
    const int row = (lane >> 4) * 4 + reg;
    lds[wave][row * 16 + (lane & 15)] = acc[reg];
  }
  __syncthreads();
  const int c = threadIdx.x;
  partials[(int64_t)blockIdx.x * 256 + c] =
      lds[0][c] + lds[1][c] + lds[2][c] + lds[3][c];
}

__global__ __launch_bounds__(THREADS) void gram_reduce_kernel(
    const float *partials, int row_chunks, const int *pair_i,
    const int *pair_j, int k, float *gram) {
  const int pair = blockIdx.x;
  const int c = threadIdx.x;
  const int row = c >> 4, col = c & 15;
  double s = 0.0;
  for (int ch = 0; ch < row_chunks; ++ch)
    s += (double)partials[((int64_t)pair * row_chunks + ch) * 256 + c];
  const int gi = pair_i[pair] * 16 + row;
  const int gj = pair_j[pair] * 16 + col;
  if (gi < k && gj < k) {
    gram[(int64_t)gi * k + gj] = (float)s;
    gram[(int64_t)gj * k + gi] = (float)s;
  }
}

extern "C" int anovos_centered_gram(const void *const *cols, int64_t n, int k,
                                    const float *means, const int *pair_i,
                                    const int *pair_j, int npairs,
                                    int row_chunks, float *partials,
                                    float *gram, hipStream_t stream) {
  dim3 grid1((uint32_t)(npairs * row_chunks));
  hipLaunchKernelGGL(gram_partials_kernel, grid1, dim3(THREADS), 0, stream,
                     (const float *const *)cols, n, k, means, pair_i, pair_j,
                     row_chunks, partials);
  hipLaunchKernelGGL(gram_reduce_kernel, dim3((uint32_t)npairs), dim3(THREADS),
                     0, stream, partials, row_chunks, pair_i, pair_j, k, gram);
  return (int)hipGetLastError();
}

// ------------------------------------------------------------------
// Single-read Gram variant. The pair-parallel kernel above re-reads
// every column once per tile-pair it appears in (~kt times, 11.7x HBM
// traffic at k=150 — measured 197 ms in the bench step). Here ONE block
// stages a 32-row slab of ALL kt*16 columns through LDS once (centered
// bf16), then its 4 waves compute EVERY 16x16 tile-pair from LDS:
// HBM traffic = n*k*4 bytes exactly.
//
// LDS layout: [4 ko-groups][ktot cols][8 rows] bf16 — one 16-byte block
// per (ko-group, col), so an MFMA A/B fragment (8 K-rows of one column)
// is a single aligned ds_read_b128; consecutive columns are 16 B apart
// (minimum-phase bank pattern for the 16 lanes of a row-group).
//
// Pairs are assigned to waves in CONTIGUOUS chunks of the i-major
// upper-tri order so a wave's consecutive pairs share the A fragment
// (compiler CSEs the identical LDS reads). Accumulators live in
// registers via a compile-time-unrolled pair loop (MAXPW per wave:
// 14 covers k<=160, 23 covers k<=208; larger k falls back).
// Deterministic: fixed block/step split, fp64 block-order reduce.
// ------------------------------------------------------------------

typedef short bf16x4_t __attribute__((ext_vector_type(4)));
typedef float nat_f4c __attribute__((ext_vector_type(4)));

// STAGES MFMA k-steps (32 rows each) are staged per barrier round: each
// column contributes STAGES*128 contiguous bytes per round (DRAM-page
// friendly — with 32-row rounds every column read was a lone 128 B
// touch on a distinct page and measured ~1.5 TB/s; 512 B touches
// amortize row activation), and barrier count drops 4x.
template <int MAXPW>
__global__ __launch_bounds__(THREADS) void gram_singleread_kernel(
    const float *const *cols, int64_t n, int k, int ktot, const float *means,
    const int *pair_i, const int *pair_j, int npairs, int row_chunks,
    float *partials) {
  constexpr int STAGES = 4;  // 128-row macro-slab
  // kstride pads each (stage, ko-group) plane by 3 column slots: the
  // un-padded plane stride (ktot*16 B = 640 words for kt=10) is 0 mod
  // 32 banks, so the 16 planes' writes all landed in the same 4 banks
  // (16-way conflict — measured 2.1 TB/s for a 75 GB pass). +3 slots
  // shifts consecutive planes by 12 banks (worst case 2-way).
  const int kstride = ktot + 3;
  extern __shared__ short slab[];  // [STAGES][4][kstride][8] bf16
  const int block = blockIdx.x;
  const int64_t macro_total = (n + STAGES * 32 - 1) / (STAGES * 32);
  const int64_t macro_per = (macro_total + row_chunks - 1) / row_chunks;
  const int64_t mac_s = (int64_t)block * macro_per;
  const int64_t mac_e = min(macro_total, mac_s + macro_per);

  const int lane = threadIdx.x & 63;
  const int wave = threadIdx.x >> 6;
  const int m = lane & 15;    // tile row (A) / tile col (B)
  const int kog = lane >> 4;  // this lane's 8-row K group

  const int ppw = (npairs + 3) / 4;  // contiguous pair chunk per wave
  const int p0 = wave * ppw;
  int i0s[MAXPW], j0s[MAXPW];
#pragma unroll
  for (int u = 0; u < MAXPW; ++u) {
    const int p = p0 + u;
    const bool act = (u < ppw) && (p < npairs);
    i0s[u] = act ? pair_i[p] * 16 : -1;
    j0s[u] = act ? pair_j[p] * 16 : 0;
  }
  f32x4 acc[MAXPW];
#pragma unroll
  for (int u = 0; u < MAXPW; ++u) acc[u] = f32x4{0.f, 0.f, 0.f, 0.f};

  for (int64_t s = mac_s; s < mac_e; ++s) {
    const int64_t r0 = s * (STAGES * 32);
    __syncthreads();  // previous round's LDS reads complete
    for (int idx = threadIdx.x; idx < ktot * 8 * STAGES; idx += THREADS) {
      const int c = idx >> (3 + 2);            // / (8*STAGES)
      const int seg = idx & (8 * STAGES - 1);  // 4-row segment in the macro-slab
      const int64_t r = r0 + (int64_t)seg * 4;
      float v0 = 0.f, v1 = 0.f, v2 = 0.f, v3 = 0.f;
      if (c < k) {
        const float *__restrict__ x = cols[c];
        const float mean = means[c];
        if (r + 4 <= n) {
          const nat_f4c t = __builtin_nontemporal_load(
              reinterpret_cast<const nat_f4c *>(x + r));
          v0 = t.x - mean; v1 = t.y - mean; v2 = t.z - mean; v3 = t.w - mean;
        } else {
          v0 = (r + 0 < n) ? x[r + 0] - mean : 0.f;
          v1 = (r + 1 < n) ? x[r + 1] - mean : 0.f;
          v2 = (r + 2 < n) ? x[r + 2] - mean : 0.f;
          v3 = (r + 3 < n) ? x[r + 3] - mean : 0.f;
        }
      }
      bf16x4_t pack;
      pack.x = to_bf16(isnan(v0) ? 0.f : v0);
      pack.y = to_bf16(isnan(v1) ? 0.f : v1);
      pack.z = to_bf16(isnan(v2) ? 0.f : v2);
      pack.w = to_bf16(isnan(v3) ? 0.f : v3);
      const int st = seg >> 3;        // stage (32-row step) 0..STAGES-1
      const int si = seg & 7;         // 4-row segment within the stage
      short *dst = slab + ((size_t)(st * 4 + (si >> 1)) * kstride + c) * 8 +
                   (size_t)(si & 1) * 4;
      *reinterpret_cast<bf16x4_t *>(dst) = pack;
    }
    __syncthreads();
#pragma unroll
    for (int st = 0; st < STAGES; ++st) {
      const short *sbase = slab + (size_t)st * 4 * kstride * 8;
#pragma unroll
      for (int u = 0; u < MAXPW; ++u) {
        if (i0s[u] >= 0) {
          const bf16x8 a = *reinterpret_cast<const bf16x8 *>(
              sbase + ((size_t)kog * kstride + i0s[u] + m) * 8);
          const bf16x8 b = *reinterpret_cast<const bf16x8 *>(
              sbase + ((size_t)kog * kstride + j0s[u] + m) * 8);
          acc[u] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a, b, acc[u], 0, 0, 0);
        }
      }
    }
  }

  // each pair is owned by exactly one wave: write fragments directly
  // (C layout: row = (lane>>4)*4 + reg, col = lane&15)
  const int row = (lane >> 4) * 4;
#pragma unroll
  for (int u = 0; u < MAXPW; ++u) {
    const int p = p0 + u;
    if (u < ppw && p < npairs) {
#pragma unroll
      for (int reg = 0; reg < 4; ++reg)
        partials[((int64_t)block * npairs + p) * 256 + (row + reg) * 16 + m] =
            acc[u][reg];
    }
  }
}

__global__ __launch_bounds__(THREADS) void gram_reduce_sr_kernel(
    const float *partials, int nblocks, int npairs, const int *pair_i,
    const int *pair_j, int k, float *gram) {
  const int pair = blockIdx.x;
  const int c = threadIdx.x;
  double s = 0.0;
  for (int ch = 0; ch < nblocks; ++ch)
    s += (double)partials[((int64_t)ch * npairs + pair) * 256 + c];
  const int row = c >> 4, col = c & 15;
  const int gi = pair_i[pair] * 16 + row;
  const int gj = pair_j[pair] * 16 + col;
  if (gi < k && gj < k) {
    gram[(int64_t)gi * k + gj] = (float)s;
    gram[(int64_t)gj * k + gi] = (float)s;
  }
}

extern "C" int anovos_centered_gram_sr(const void *const *cols, int64_t n,
                                       int k, const float *means,
                                       const int *pair_i, const int *pair_j,
                                       int npairs, int row_chunks,
                                       float *partials, float *gram,
                                       hipStream_t stream) {
  const int kt = (k + 15) / 16;
  const int ktot = kt * 16;
  const size_t lds = (size_t)4 /*STAGES*/ * 4 * (ktot + 3) * 16;
  if (kt <= 10)
    hipLaunchKernelGGL((gram_singleread_kernel<14>), dim3(row_chunks),
                       dim3(THREADS), lds, stream, (const float *const *)cols,
                       n, k, ktot, means, pair_i, pair_j, npairs, row_chunks,
                       partials);
  else if (kt <= 13)
    hipLaunchKernelGGL((gram_singleread_kernel<23>), dim3(row_chunks),
                       dim3(THREADS), lds, stream, (const float *const *)cols,
                       n, k, ktot, means, pair_i, pair_j, npairs, row_chunks,
                       partials);
  else
    return -2;  // caller dispatches the pair-parallel kernel instead
  hipLaunchKernelGGL(gram_reduce_sr_kernel, dim3((uint32_t)npairs),
                     dim3(THREADS), 0, stream, partials, row_chunks, npairs,
                     pair_i, pair_j, k, gram);
  return (int)hipGetLastError();
}

extern "C" int anovos_gram_sr_grid(int k) {
  // Preferred grid = EXACTLY the resident block capacity (one wave of
  // blocks): each block then owns the longest possible contiguous
  // per-column row range, which is what the 150-stream-per-block access
  // pattern needs for DRAM locality (sweep at 20M x 150: 256 blocks
  // 0.95 TB/s, 768 = capacity 2.20 TB/s, 2048 1.93, 16384 1.01).
  const int kt = (k + 15) / 16;
  const int ktot = kt * 16;
  const size_t lds = (size_t)4 * 4 * (ktot + 3) * 16;
  static int cached_k = -1, cached_grid = 0;
  if (k == cached_k) return cached_grid;
  int per_cu = 0;
  using KFn = void (*)(const float *const *, int64_t, int, int, const float *,
                       const int *, const int *, int, int, float *);
  KFn fn = (kt <= 10) ? (KFn)gram_singleread_kernel<14>
                      : (KFn)gram_singleread_kernel<23>;
  (void)hipOccupancyMaxActiveBlocksPerMultiprocessor(
      &per_cu, reinterpret_cast<const void *>(fn), THREADS, lds);
  if (per_cu <= 0) per_cu = 3;
  hipDeviceProp_t prop;
  int dev = 0;
  hipGetDevice(&dev);
  hipGetDeviceProperties(&prop, dev);
  int grid = per_cu * (prop.multiProcessorCount > 0 ? prop.multiProcessorCount : 256);
  cached_k = k;
  cached_grid = grid;
  return grid;
}
