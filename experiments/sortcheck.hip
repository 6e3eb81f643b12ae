// Isolated rocPRIM radix_sort_pairs + reduce_by_key integrity check at the
// louvain-coarsen scale (experiment tooling; r02 storm bisect).
#include <cstring>

#include <hip/hip_runtime.h>
#include <rocprim/rocprim.hpp>
#include <cstdio>
#include <cstdint>

#define TRY(x) do { hipError_t e=(x); if(e!=hipSuccess){printf("ERR %s:%d %s\n",__FILE__,__LINE__,hipGetErrorString(e)); return 1;} } while(0)

__global__ void k_gen(int64_t n, uint64_t ncl, uint64_t *keys, double *vals) {
  for (int64_t i = blockIdx.x*(int64_t)blockDim.x+threadIdx.x; i < n;
       i += (int64_t)gridDim.x*blockDim.x) {
    uint64_t x = (uint64_t)i * 0x9E3779B97F4A7C15ull;
    x = (x ^ (x>>30)) * 0xBF58476D1CE4E5B9ull; x ^= x>>31;
    uint64_t ci = x % ncl, ct = (x >> 21) % ncl;
    if (ci < ct) { keys[i] = ~0ull; vals[i] = 0.0; }
    else { keys[i] = (ci<<32)|ct; vals[i] = 1.0; }
  }
}

__global__ void k_check(int64_t n, const uint64_t *keys, uint64_t ncl,
                        unsigned long long *bad_range, unsigned long long *bad_order) {
  unsigned long long br=0, bo=0;
  for (int64_t i = blockIdx.x*(int64_t)blockDim.x+threadIdx.x; i < n;
       i += (int64_t)gridDim.x*blockDim.x) {
    uint64_t k = keys[i];
    if (k != ~0ull) {
      uint64_t ci = k>>32, ct = (uint32_t)k;
      if (ci >= ncl || ct >= ncl || ct > ci) ++br;
    }
    if (i+1 < n && keys[i] > keys[i+1]) ++bo;
  }
  if (br) atomicAdd(bad_range, br);
  if (bo) atomicAdd(bad_order, bo);
}

int main(int argc, char **argv) {
  const int64_t n = argc > 1 ? atoll(argv[1]) : 536870912;
  const uint64_t ncl = 3106911;
  uint64_t *keys, *keys_out, *ukeys; double *vals, *vals_out, *uvals;
  unsigned int *ucount; unsigned long long *bad;
  TRY(hipMalloc(&keys, n*8)); TRY(hipMalloc(&keys_out, n*8));
  TRY(hipMalloc(&vals, n*8)); TRY(hipMalloc(&vals_out, n*8));
  TRY(hipMalloc(&ukeys, n*8)); TRY(hipMalloc(&uvals, n*8));
  TRY(hipMalloc(&ucount, 4)); TRY(hipMalloc(&bad, 16));
  for (int rep = 0; rep < 5; ++rep) {
    hipLaunchKernelGGL(k_gen, dim3(4096), dim3(256), 0, 0, n, ncl, keys, vals);
    TRY(hipMemset(bad, 0, 16));
    hipLaunchKernelGGL(k_check, dim3(4096), dim3(256), 0, 0, n, keys, ncl, bad, bad+1);
    unsigned long long h0[2]; TRY(hipMemcpy(h0, bad, 16, hipMemcpyDeviceToHost));
    size_t tmp_bytes = 0; void *tmp = nullptr;
    TRY(rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys, keys_out, vals, vals_out, n, 0, 64));
    TRY(hipMalloc(&tmp, tmp_bytes));
    TRY(rocprim::radix_sort_pairs(tmp, tmp_bytes, keys, keys_out, vals, vals_out, n, 0, 64));
    TRY(hipMemset(bad, 0, 16));
    hipLaunchKernelGGL(k_check, dim3(4096), dim3(256), 0, 0, n, keys_out, ncl, bad, bad+1);
    unsigned long long h1[2]; TRY(hipMemcpy(h1, bad, 16, hipMemcpyDeviceToHost));
    size_t tmp2 = 0; void *t2 = nullptr;
    TRY(rocprim::reduce_by_key(nullptr, tmp2, keys_out, vals_out, n, ukeys, uvals, ucount,
                               rocprim::plus<double>(), rocprim::equal_to<uint64_t>()));
    TRY(hipMalloc(&t2, tmp2));
    TRY(rocprim::reduce_by_key(t2, tmp2, keys_out, vals_out, n, ukeys, uvals, ucount,
                               rocprim::plus<double>(), rocprim::equal_to<uint64_t>()));
    unsigned int nu = 0; TRY(hipMemcpy(&nu, ucount, 4, hipMemcpyDeviceToHost));
    TRY(hipMemset(bad, 0, 16));
    hipLaunchKernelGGL(k_check, dim3(4096), dim3(256), 0, 0, (int64_t)nu, ukeys, ncl, bad, bad+1);
    unsigned long long h2[2]; TRY(hipMemcpy(h2, bad, 16, hipMemcpyDeviceToHost));
    printf("rep %d: pre bad=%llu | post-sort bad=%llu disorder=%llu | nu=%u ukeys bad=%llu\n",
           rep, h0[0], h1[0], h1[1], nu, h2[0]);
    TRY(hipFree(tmp)); TRY(hipFree(t2));
  }
  return 0;
}
