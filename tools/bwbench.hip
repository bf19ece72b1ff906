// Streaming-bandwidth microbenchmark for the engine's out-of-place
// column kernels (fillnan/axpb/bucketize/clamp shapes). Measures
// effective read+write TB/s for several loop shapes on gfx950 so the
// production kernels copy at the measured ceiling (~6.3 TB/s float4
// copy per MI355X_MICROARCH.md) instead of ~4.5.
//
// Build: hipcc --offload-arch=gfx950 -O3 tools/bwbench.hip -o gpurun_out/bwbench
// Run:   ./gpurun_out/bwbench

#include <hip/hip_runtime.h>

#include <cstdio>
#include <cstdint>

#define CHECK(x)                                                              \
  do {                                                                        \
    hipError_t e = (x);                                                       \
    if (e != hipSuccess) {                                                    \
      printf("HIP error %s at %d\n", hipGetErrorString(e), __LINE__);         \
      return 1;                                                               \
    }                                                                         \
  } while (0)

constexpr int THREADS = 256;

// v1: the current production shape — per-block contiguous span, one
// float4 per lane per iteration
__global__ __launch_bounds__(THREADS) void k_copy_v1(const float *__restrict__ x,
                                                     float *__restrict__ out,
                                                     int64_t n, int nchunks) {
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const float4 *xv = reinterpret_cast<const float4 *>(x + s);
  float4 *ov = reinterpret_cast<float4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  for (int64_t i = threadIdx.x; i < nv; i += THREADS) ov[i] = xv[i];
}

// v2: two float4 in flight per lane (explicit 2x unroll)
__global__ __launch_bounds__(THREADS) void k_copy_v2(const float *__restrict__ x,
                                                     float *__restrict__ out,
                                                     int64_t n, int nchunks) {
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const float4 *__restrict__ xv = reinterpret_cast<const float4 *>(x + s);
  float4 *__restrict__ ov = reinterpret_cast<float4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  int64_t i = threadIdx.x;
  for (; i + THREADS < nv; i += 2 * THREADS) {
    float4 a = xv[i];
    float4 b = xv[i + THREADS];
    ov[i] = a;
    ov[i + THREADS] = b;
  }
  for (; i < nv; i += THREADS) ov[i] = xv[i];
}

// v4: four float4 in flight per lane
__global__ __launch_bounds__(THREADS) void k_copy_v4(const float *__restrict__ x,
                                                     float *__restrict__ out,
                                                     int64_t n, int nchunks) {
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const float4 *__restrict__ xv = reinterpret_cast<const float4 *>(x + s);
  float4 *__restrict__ ov = reinterpret_cast<float4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  int64_t i = threadIdx.x;
  for (; i + 3 * THREADS < nv; i += 4 * THREADS) {
    float4 a = xv[i];
    float4 b = xv[i + THREADS];
    float4 c = xv[i + 2 * THREADS];
    float4 d = xv[i + 3 * THREADS];
    ov[i] = a;
    ov[i + THREADS] = b;
    ov[i + 2 * THREADS] = c;
    ov[i + 3 * THREADS] = d;
  }
  for (; i < nv; i += THREADS) ov[i] = xv[i];
}

// v2nt: 2x unroll + non-temporal load/store (needs the native ext
// vector type, HIP_vector_type is not accepted by the builtin)
typedef float nat_f4 __attribute__((ext_vector_type(4)));

__global__ __launch_bounds__(THREADS) void k_copy_v2nt(const float *__restrict__ x,
                                                       float *__restrict__ out,
                                                       int64_t n, int nchunks) {
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const nat_f4 *__restrict__ xv = reinterpret_cast<const nat_f4 *>(x + s);
  nat_f4 *__restrict__ ov = reinterpret_cast<nat_f4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  int64_t i = threadIdx.x;
  for (; i + THREADS < nv; i += 2 * THREADS) {
    nat_f4 a = __builtin_nontemporal_load(&xv[i]);
    nat_f4 b = __builtin_nontemporal_load(&xv[i + THREADS]);
    __builtin_nontemporal_store(a, &ov[i]);
    __builtin_nontemporal_store(b, &ov[i + THREADS]);
  }
  for (; i < nv; i += THREADS) {
    nat_f4 a = xv[i];
    ov[i] = a;
  }
}

// v2 with the fillnan select, to price the ALU work
__global__ __launch_bounds__(THREADS) void k_fill_v2(const float *__restrict__ x,
                                                     float *__restrict__ out,
                                                     int64_t n, int nchunks) {
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const float4 *__restrict__ xv = reinterpret_cast<const float4 *>(x + s);
  float4 *__restrict__ ov = reinterpret_cast<float4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  const float ff = 0.5f;
  int64_t i = threadIdx.x;
  for (; i + THREADS < nv; i += 2 * THREADS) {
    float4 a = xv[i];
    float4 b = xv[i + THREADS];
    a.x = isnan(a.x) ? ff : a.x; a.y = isnan(a.y) ? ff : a.y;
    a.z = isnan(a.z) ? ff : a.z; a.w = isnan(a.w) ? ff : a.w;
    b.x = isnan(b.x) ? ff : b.x; b.y = isnan(b.y) ? ff : b.y;
    b.z = isnan(b.z) ? ff : b.z; b.w = isnan(b.w) ? ff : b.w;
    ov[i] = a;
    ov[i + THREADS] = b;
  }
  for (; i < nv; i += THREADS) ov[i] = xv[i];
}

// v1 at 1024 threads per block
__global__ __launch_bounds__(1024) void k_copy_t1024(const float *__restrict__ x,
                                                     float *__restrict__ out,
                                                     int64_t n, int nchunks) {
  const int chunk = blockIdx.x;
  const int64_t per = (n + nchunks - 1) / nchunks;
  const int64_t s = (int64_t)chunk * per;
  const int64_t e = min(n, s + per);
  const float4 *__restrict__ xv = reinterpret_cast<const float4 *>(x + s);
  float4 *__restrict__ ov = reinterpret_cast<float4 *>(out + s);
  const int64_t nv = (e - s) / 4;
  for (int64_t i = threadIdx.x; i < nv; i += 1024) ov[i] = xv[i];
}

// flat grid-stride (no chunk spans): float4 index = global thread stride
__global__ __launch_bounds__(THREADS) void k_copy_gridstride(const float *__restrict__ x,
                                                             float *__restrict__ out,
                                                             int64_t n, int nchunks) {
  const float4 *__restrict__ xv = reinterpret_cast<const float4 *>(x);
  float4 *__restrict__ ov = reinterpret_cast<float4 *>(out);
  const int64_t nv = n / 4;
  const int64_t stride = (int64_t)gridDim.x * THREADS;
  for (int64_t i = (int64_t)blockIdx.x * THREADS + threadIdx.x; i < nv; i += stride)
    ov[i] = xv[i];
}

template <typename K>
double bench(K kern, const float *x, float *out, int64_t n, int nchunks,
             int threads, const char *name) {
  hipEvent_t a, b;
  hipEventCreate(&a);
  hipEventCreate(&b);
  // warmup
  hipLaunchKernelGGL(kern, dim3(nchunks), dim3(threads), 0, 0, x, out, n, nchunks);
  hipDeviceSynchronize();
  hipEventRecord(a);
  for (int r = 0; r < 5; ++r)
    hipLaunchKernelGGL(kern, dim3(nchunks), dim3(threads), 0, 0, x, out, n, nchunks);
  hipEventRecord(b);
  hipDeviceSynchronize();
  float ms = 0;
  hipEventElapsedTime(&ms, a, b);
  double tbps = 2.0 * n * 4 * 5 / (ms / 1000.0) / 1e12;
  printf("%-16s %7.3f ms/iter  %6.2f TB/s\n", name, ms / 5, tbps);
  hipEventDestroy(a);
  hipEventDestroy(b);
  return tbps;
}

int main() {
  const int64_t n = 1LL << 30;  // 4 GB in, 4 GB out
  float *x, *out;
  CHECK(hipMalloc(&x, n * 4));
  CHECK(hipMalloc(&out, n * 4));
  CHECK(hipMemset(x, 0x3f, n * 4));
  for (int nchunks : {2048, 4096, 16384}) {
    printf("--- nchunks=%d ---\n", nchunks);
    bench(k_copy_v1, x, out, n, nchunks, THREADS, "v1_current");
    bench(k_copy_v2, x, out, n, nchunks, THREADS, "v2_unroll2");
    bench(k_copy_v4, x, out, n, nchunks, THREADS, "v4_unroll4");
    bench(k_copy_v2nt, x, out, n, nchunks, THREADS, "v2_nontemporal");
    bench(k_fill_v2, x, out, n, nchunks, THREADS, "fill_v2");
    bench(k_copy_t1024, x, out, n, nchunks, 1024, "v1_1024thr");
    bench(k_copy_gridstride, x, out, n, nchunks, THREADS, "gridstride");
  }
  return 0;
}
