// Gather-bandwidth microbenchmark for MI355X (experiment tooling).
//
// Measures the chip's achievable rate for the PageRank sweep's exact access
// shape — random 4-B gathers from an N-byte f32 table driven by a streamed
// index array — as a function of table size (4 MB..1 GB) and gathers in
// flight per lane. This bounds what ANY vertex ordering can achieve and
// decides whether the sweep is traffic-bound or issue/latency-bound
// (DESIGN.md round-2 item 1).
//
// Build: hipcc --offload-arch=gfx950 -O3 gatherbench.hip -o gatherbench
// Run:   ./gatherbench [table_mb...]
#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <vector>

#define TRY(x)                                                   \
  do {                                                           \
    hipError_t e = (x);                                          \
    if (e != hipSuccess) {                                       \
      fprintf(stderr, "%s:%d %s\n", __FILE__, __LINE__,          \
              hipGetErrorString(e));                             \
      exit(1);                                                   \
    }                                                            \
  } while (0)

constexpr int kBlock = 256;

__global__ void k_fill_idx(int64_t n, uint32_t mask, uint64_t seed, int32_t *idx) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t x = seed + (uint64_t)i * 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    x ^= x >> 31;
    idx[i] = (int32_t)((uint32_t)x & mask);
  }
}

// The sweep's wide-row shape: int4 nontemporal index loads + G independent
// gathers per lane in flight, f64 accumulation.
template <int UNROLL>
__global__ void __launch_bounds__(kBlock) k_gather(int64_t n, const int32_t *idx,
                                                   const float *table, double *out) {
  typedef int v4i __attribute__((ext_vector_type(4)));
  const v4i *idx4 = reinterpret_cast<const v4i *>(idx);
  const int64_t n4 = n / 4;
  double acc = 0.0;
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x;
  for (; i + (UNROLL - 1) * stride < n4; i += UNROLL * stride) {
    float g[4 * UNROLL];
#pragma unroll
    for (int u = 0; u < UNROLL; ++u) {
      const v4i c = __builtin_nontemporal_load(idx4 + i + u * stride);
      g[4 * u + 0] = table[c.x];
      g[4 * u + 1] = table[c.y];
      g[4 * u + 2] = table[c.z];
      g[4 * u + 3] = table[c.w];
    }
#pragma unroll
    for (int k = 0; k < 4 * UNROLL; ++k) acc += (double)g[k];
  }
  for (; i < n4; i += stride) {
    const v4i c = __builtin_nontemporal_load(idx4 + i);
    acc += (double)table[c.x] + (double)table[c.y] + (double)table[c.z] +
           (double)table[c.w];
  }
  __shared__ double red[kBlock / 64];
  for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) atomicAdd(out, red[0] + red[1] + red[2] + red[3]);
}

template <int UNROLL>
double run_one(int64_t n_idx, const int32_t *d_idx, const float *d_table, double *d_out,
               int grid) {
  hipEvent_t e0, e1;
  TRY(hipEventCreate(&e0));
  TRY(hipEventCreate(&e1));
  // warmup
  hipLaunchKernelGGL((k_gather<UNROLL>), dim3(grid), dim3(kBlock), 0, 0, n_idx, d_idx,
                     d_table, d_out);
  TRY(hipDeviceSynchronize());
  TRY(hipEventRecord(e0));
  for (int r = 0; r < 5; ++r)
    hipLaunchKernelGGL((k_gather<UNROLL>), dim3(grid), dim3(kBlock), 0, 0, n_idx, d_idx,
                       d_table, d_out);
  TRY(hipEventRecord(e1));
  TRY(hipEventSynchronize(e1));
  float ms = 0;
  TRY(hipEventElapsedTime(&ms, e0, e1));
  TRY(hipEventDestroy(e0));
  TRY(hipEventDestroy(e1));
  return ms / 5.0;
}

int main(int argc, char **argv) {
  const int64_t n_idx = 1ll << 30;  // 1Gi gathers per pass (the RMAT-26 edge count)
  int32_t *d_idx = nullptr;
  double *d_out = nullptr;
  TRY(hipMalloc(&d_idx, n_idx * 4));
  TRY(hipMalloc(&d_out, 8));
  std::vector<long> sizes = {4, 16, 32, 64, 108, 128, 256, 512, 1024};
  if (argc > 1) {
    sizes.clear();
    for (int a = 1; a < argc; ++a) sizes.push_back(atol(argv[a]));
  }
  printf("gathers=%lld per pass; effective bytes/gather = 4 (idx) + 4 (val)\n",
         (long long)n_idx);
  printf("%8s %6s %6s %10s %12s %12s\n", "table_mb", "unroll", "grid", "ms",
         "Ggather/s", "eff_GB/s");
  for (long mb : sizes) {
    // table sizes are powers of two for masking; 108 -> 128-mask truncated
    uint32_t entries = (uint32_t)((mb << 20) / 4);
    uint32_t mask = 1;
    while ((mask << 1) <= entries) mask <<= 1;
    mask -= 1;  // gathers within the largest pow2 <= size
    float *d_table = nullptr;
    TRY(hipMalloc(&d_table, (size_t)(mask + 1) * 4));
    TRY(hipMemset(d_table, 0x3f, (size_t)(mask + 1) * 4));
    hipLaunchKernelGGL(k_fill_idx, dim3(4096), dim3(kBlock), 0, 0, n_idx, mask, 1ull,
                       d_idx);
    TRY(hipDeviceSynchronize());
    for (int grid : {8192}) {
      double ms2 = run_one<2>(n_idx, d_idx, d_table, d_out, grid);
      double ms4 = run_one<4>(n_idx, d_idx, d_table, d_out, grid);
      double ms8 = run_one<8>(n_idx, d_idx, d_table, d_out, grid);
      for (auto [u, ms] : {std::pair<int, double>{2, ms2}, {4, ms4}, {8, ms8}}) {
        const double gps = (double)n_idx / (ms * 1e6);
        printf("%8ld %6d %6d %10.3f %12.2f %12.1f\n", mb, u, grid, ms, gps, gps * 8);
      }
    }
    TRY(hipFree(d_table));
  }
  TRY(hipFree(d_idx));
  TRY(hipFree(d_out));
  return 0;
}
