// Device-side graph construction for gfx950: deterministic edge-stream
// generation, COO -> CSR (histogram + exclusive scan + scatter), and
// degree-bin work lists for the fused sweep kernels.
//
// Replaces (MI355X-native, from scratch) the host-side layouts built by
// pagerank_alg::PageRankGraph (reference algorithm/pagerank.cpp:166-182) and
// GetGrappoloSuitableGraph (reference louvain.cpp:158-233; sym-CSR, every
// edge stored twice) with int32 indices / fp32 weights in HBM.

#include <cstring>

#include <rocprim/rocprim.hpp>

#include "../../include/mgx_graphgen.h"
#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  if (g > cap) g = cap;
  return g;
}

__global__ void k_rmat(int64_t n_edges, int scale, uint64_t mixed_seed,
                       mgx_rmat_thresholds t, int32_t *src, int32_t *dst) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t s, d;
    mgx_rmat_edge(mixed_seed, (uint64_t)i, scale, t, &s, &d);
    src[i] = (int32_t)s;
    dst[i] = (int32_t)d;
  }
}

__global__ void k_uniform(int64_t n_edges, uint64_t mixed_seed, uint64_t n_vertices,
                          int32_t *src, int32_t *dst) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    uint64_t s, d;
    mgx_uniform_edge(mixed_seed, (uint64_t)i, n_vertices, &s, &d);
    src[i] = (int32_t)s;
    dst[i] = (int32_t)d;
  }
}

__global__ void k_weights(int64_t n_edges, uint64_t mixed_seed, float *w) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    w[i] = (float)mgx_edge_weight(mixed_seed, (uint64_t)i);
  }
}

__global__ void k_hist(int64_t n_edges, const int32_t *idx, uint32_t *counts) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x)
    atomicAdd(&counts[idx[i]], 1u);
}

__global__ void k_hist2(int64_t n_edges, const int32_t *a, const int32_t *b,
                        uint32_t *counts) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    atomicAdd(&counts[a[i]], 1u);
    atomicAdd(&counts[b[i]], 1u);
  }
}

__global__ void k_hist_ranged(int64_t n_edges, const int32_t *idx, int32_t lo, int32_t hi,
                              uint32_t *counts /* [hi-lo] */) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    int32_t v = idx[i];
    if (v >= lo && v < hi) atomicAdd(&counts[v - lo], 1u);
  }
}

__global__ void k_iota_i32g(int64_t n, int32_t *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = (int32_t)i;
}

__global__ void k_inv_outdeg(int64_t n, const uint32_t *deg, float *inv) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    inv[i] = deg[i] ? 1.0f / (float)deg[i] : 0.0f;
}

__global__ void k_bin_keys(int64_t rows, const uint32_t *lo, const uint32_t *hi,
                           int include_zero, uint32_t *keys, uint32_t *vals) {
  for (int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; r < rows;
       r += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t deg = hi[r] - lo[r];
    uint32_t k = deg < 8 ? 0u : deg < 64 ? 1u : deg < 1024 ? 2u : 3u;
    if (deg == 0 && !include_zero) k = 4u;  // dropped (sorts last)
    keys[r] = k;
    vals[r] = (uint32_t)r;
  }
}

__global__ void k_count_bins(int64_t rows, const uint32_t *keys, uint32_t *counts5) {
  // Per-block LDS aggregation first: a naive global histogram measured
  // 706 ms at RMAT-26 (atomic hotspot); this form is ~ms. Key 4 counts the
  // dropped zero-degree rows.
  __shared__ uint32_t local[5];
  if (threadIdx.x < 5) local[threadIdx.x] = 0;
  __syncthreads();
  for (int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; r < rows;
       r += (int64_t)gridDim.x * blockDim.x)
    atomicAdd(&local[keys[r]], 1u);
  __syncthreads();
  if (threadIdx.x < 5 && local[threadIdx.x]) atomicAdd(&counts5[threadIdx.x], local[threadIdx.x]);
}

__global__ void k_pack_pairs(int64_t n_edges, const int32_t *src, const int32_t *dst,
                             const int32_t *perm, uint64_t *keys) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t d = perm ? (uint32_t)perm[dst[i]] : (uint32_t)dst[i];
    const uint32_t sr = perm ? (uint32_t)perm[src[i]] : (uint32_t)src[i];
    keys[i] = ((uint64_t)d << 32) | sr;
  }
}

__global__ void k_invert_perm(int64_t n, const int32_t *order, int32_t *perm) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    perm[order[i]] = (int32_t)i;
}

__global__ void k_gather_u32(int64_t n, const uint32_t *in, const int32_t *order,
                             uint32_t *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = in[order[i]];
}

__global__ void k_hist_perm(int64_t n_edges, const int32_t *idx, const int32_t *perm,
                            uint32_t *counts) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x)
    atomicAdd(&counts[perm[idx[i]]], 1u);
}

__global__ void k_pack_pairs_ranged(int64_t n_edges, const int32_t *src, const int32_t *dst,
                                    int32_t lo, int32_t hi, uint64_t *keys) {
  // Out-of-range edges get the max key: they sort last and are trimmed.
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t d = dst[i];
    keys[i] = (d >= lo && d < hi)
                  ? ((uint64_t)(uint32_t)(d - lo) << 32) | (uint32_t)src[i]
                  : ~0ull;
  }
}

__global__ void k_unpack_cols(int64_t n, const uint64_t *keys, int32_t *col) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    col[i] = (int32_t)(uint32_t)keys[i];
}

// Build a sorted in-CSR column array: cols within each row in ascending
// order (gather locality: adjacent cols share cache lines — the measured
// unsorted-scatter build over-fetched 3x algorithmic bytes in the sweep).
mgx_status build_sorted_cols(mgx_context *ctx, const int32_t *d_src, const int32_t *d_dst,
                             const int32_t *perm, int64_t n_edges, int64_t key_rows,
                             bool ranged, int32_t lo, int32_t hi, int32_t *col,
                             int64_t col_count) {
  if (n_edges == 0) return MGX_OK;
  uint64_t *keys = nullptr, *keys_out = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&keys, n_edges * sizeof(uint64_t)));
  MGX_TRY(ctx->alloc_async((void **)&keys_out, n_edges * sizeof(uint64_t)));
  if (ranged) {
    hipLaunchKernelGGL(k_pack_pairs_ranged, dim3(grid_for(n_edges)), dim3(kBlock), 0,
                       ctx->stream, n_edges, d_src, d_dst, lo, hi, keys);
  } else {
    hipLaunchKernelGGL(k_pack_pairs, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                       n_edges, d_src, d_dst, perm, keys);
  }
  int end_bit = 33;
  while ((1ll << (end_bit - 32)) < key_rows + 1) ++end_bit;
  if (ranged) end_bit = 64;  // the ~0 sentinel must sort last
  size_t tmp_bytes = 0;
  auto err = rocprim::radix_sort_keys(nullptr, tmp_bytes, keys, keys_out, n_edges, 0,
                                      end_bit, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::radix_sort_keys(tmp, tmp_bytes, keys, keys_out, n_edges, 0, end_bit,
                                 ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  hipLaunchKernelGGL(k_unpack_cols, dim3(grid_for(col_count)), dim3(kBlock), 0, ctx->stream,
                     col_count, keys_out, col);
  MGX_TRY(ctx->free_async(keys));
  MGX_TRY(ctx->free_async(keys_out));
  return MGX_OK;
}

// Weighted variant of build_sorted_cols (identity layout only): carries the
// fp32 edge weight through the (dst<<32|src) sort so in_w[j] matches
// in_col[j]. Keys sort on the full 64 bits so weight order is deterministic
// (ties between parallel edges keep ascending weight-payload order — the
// weights of parallel edges are interchangeable for every consumer anyway).
mgx_status build_sorted_cols_w(mgx_context *ctx, const int32_t *d_src,
                               const int32_t *d_dst, const float *d_w, int64_t n_edges,
                               int64_t key_rows, int32_t *col, float *out_w) {
  if (n_edges == 0) return MGX_OK;
  uint64_t *keys = nullptr, *keys_out = nullptr;
  float *vals_out = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&keys, n_edges * sizeof(uint64_t)));
  MGX_TRY(ctx->alloc_async((void **)&keys_out, n_edges * sizeof(uint64_t)));
  MGX_TRY(ctx->alloc_async((void **)&vals_out, n_edges * sizeof(float)));
  hipLaunchKernelGGL(k_pack_pairs, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                     n_edges, d_src, d_dst, (const int32_t *)nullptr, keys);
  int end_bit = 33;
  while ((1ll << (end_bit - 32)) < key_rows + 1) ++end_bit;
  size_t tmp_bytes = 0;
  auto err = rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys, keys_out,
                                       (const float *)d_w, vals_out, n_edges, 0, end_bit,
                                       ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::radix_sort_pairs(tmp, tmp_bytes, keys, keys_out, (const float *)d_w,
                                  vals_out, n_edges, 0, end_bit, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  hipLaunchKernelGGL(k_unpack_cols, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                     n_edges, keys_out, col);
  MGX_HIP_TRY(hipMemcpyAsync(out_w, vals_out, n_edges * sizeof(float),
                             hipMemcpyDeviceToDevice, ctx->stream));
  MGX_TRY(ctx->free_async(keys));
  MGX_TRY(ctx->free_async(keys_out));
  MGX_TRY(ctx->free_async(vals_out));
  return MGX_OK;
}

__global__ void k_pack_sym(int64_t n_edges, const int32_t *src, const int32_t *dst,
                           const float *w, uint64_t *keys, float *vals) {
  // Each input edge twice: (s->d) and (d->s), the GetGrappoloSuitableGraph
  // layout (louvain.cpp:176-233), as sortable (row<<32|col) keys.
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_edges;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t a = (uint32_t)src[i], b = (uint32_t)dst[i];
    const float wi = w ? w[i] : 1.0f;
    keys[2 * i] = ((uint64_t)a << 32) | b;
    keys[2 * i + 1] = ((uint64_t)b << 32) | a;
    if (vals) {
      vals[2 * i] = wi;
      vals[2 * i + 1] = wi;
    }
  }
}

// Sorted symmetric CSR (cols ascending within each row) via pair sort.
mgx_status build_sorted_sym(mgx_context *ctx, const int32_t *d_src, const int32_t *d_dst,
                            const float *d_w, int64_t n_edges, int64_t n_vertices,
                            int32_t *col, float *out_w) {
  if (n_edges == 0) return MGX_OK;
  const int64_t n2 = 2 * n_edges;
  uint64_t *keys = nullptr, *keys_out = nullptr;
  float *vals = nullptr, *vals_out = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&keys, n2 * sizeof(uint64_t)));
  MGX_TRY(ctx->alloc_async((void **)&keys_out, n2 * sizeof(uint64_t)));
  if (out_w) {
    MGX_TRY(ctx->alloc_async((void **)&vals, n2 * sizeof(float)));
    MGX_TRY(ctx->alloc_async((void **)&vals_out, n2 * sizeof(float)));
  }
  hipLaunchKernelGGL(k_pack_sym, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                     n_edges, d_src, d_dst, d_w, keys, vals);
  int end_bit = 33;
  while ((1ll << (end_bit - 32)) < n_vertices + 1) ++end_bit;
  size_t tmp_bytes = 0;
  hipError_t err;
  if (out_w) {
    err = rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys, keys_out, vals, vals_out, n2,
                                    0, end_bit, ctx->stream);
  } else {
    err = rocprim::radix_sort_keys(nullptr, tmp_bytes, keys, keys_out, n2, 0, end_bit,
                                   ctx->stream);
  }
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  if (out_w) {
    err = rocprim::radix_sort_pairs(tmp, tmp_bytes, keys, keys_out, vals, vals_out, n2, 0,
                                    end_bit, ctx->stream);
  } else {
    err = rocprim::radix_sort_keys(tmp, tmp_bytes, keys, keys_out, n2, 0, end_bit,
                                   ctx->stream);
  }
  if (err != hipSuccess) return MGX_ERR_HIP;
  hipLaunchKernelGGL(k_unpack_cols, dim3(grid_for(n2)), dim3(kBlock), 0, ctx->stream, n2,
                     keys_out, col);
  if (out_w) {
    MGX_HIP_TRY(hipMemcpyAsync(out_w, vals_out, n2 * sizeof(float),
                               hipMemcpyDeviceToDevice, ctx->stream));
  }
  MGX_TRY(ctx->free_async(keys));
  MGX_TRY(ctx->free_async(keys_out));
  if (vals) MGX_TRY(ctx->free_async(vals));
  if (vals_out) MGX_TRY(ctx->free_async(vals_out));
  return MGX_OK;
}

__global__ void k_i64_to_i32(int64_t n, const int64_t *in, int32_t *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int32_t)in[i];
}

__global__ void k_f64_to_f32(int64_t n, const double *in, float *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (float)in[i];
}

// Exclusive scan of u32 counts[n] -> row_ptr[n+1] (row_ptr[n] = total).
mgx_status scan_counts(mgx_context *ctx, const uint32_t *counts, int64_t n,
                       uint32_t *row_ptr) {
  size_t tmp_bytes = 0;
  auto err = rocprim::exclusive_scan(nullptr, tmp_bytes, counts, row_ptr, 0u, n + 1,
                                     rocprim::plus<uint32_t>(), ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  // Scan n+1 inputs (counts has n entries; read one past would be invalid) —
  // instead scan n entries into row_ptr[0..n) and set the total separately.
  err = rocprim::exclusive_scan(tmp, tmp_bytes, counts, row_ptr, 0u, n,
                                rocprim::plus<uint32_t>(), ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  // row_ptr[n] = row_ptr[n-1] + counts[n-1]
  uint32_t last_off = 0, last_cnt = 0;
  if (n > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(&last_off, row_ptr + n - 1, 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(&last_cnt, counts + n - 1, 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }
  const uint32_t total = last_off + last_cnt;
  MGX_HIP_TRY(hipMemcpyAsync(row_ptr + n, &total, 4, hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  return MGX_OK;
}

}  // namespace

mgx_status mgx_gen_rmat_device(mgx_context *ctx, int scale, int64_t n_edges, uint64_t seed,
                               double a, double b, double c, int32_t *d_src, int32_t *d_dst) {
  const uint64_t ms = mgx_seed_mix(seed);
  const mgx_rmat_thresholds t = mgx_rmat_make_thresholds(a, b, c);
  hipLaunchKernelGGL(k_rmat, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream, n_edges,
                     scale, ms, t, d_src, d_dst);
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}

mgx_status mgx_gen_uniform_device(mgx_context *ctx, int64_t n_vertices, int64_t n_edges,
                                  uint64_t seed, int32_t *d_src, int32_t *d_dst) {
  const uint64_t ms = mgx_seed_mix(seed);
  hipLaunchKernelGGL(k_uniform, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                     n_edges, ms, (uint64_t)n_vertices, d_src, d_dst);
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}

mgx_status mgx_gen_weights_device(mgx_context *ctx, int64_t n_edges, uint64_t seed,
                                  float *d_w) {
  const uint64_t ms = mgx_seed_mix(seed);
  hipLaunchKernelGGL(k_weights, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                     n_edges, ms, d_w);
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}

mgx_status mgx_build_bins_range(mgx_context *ctx, const uint32_t *lo, const uint32_t *hi,
                                int64_t rows, bool include_zero, mgx_bins *bins) {
  if (rows == 0) {
    bins->rows = nullptr;
    return MGX_OK;
  }
  MGX_HIP_TRY(mgx_hip_malloc(&bins->rows, rows * sizeof(int32_t)));
  uint32_t *keys = nullptr, *keys_out = nullptr, *vals = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&keys, rows * sizeof(uint32_t)));
  MGX_HIP_TRY(mgx_hip_malloc(&keys_out, rows * sizeof(uint32_t)));
  MGX_HIP_TRY(mgx_hip_malloc(&vals, rows * sizeof(uint32_t)));
  uint32_t *counts5 = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&counts5, 5 * sizeof(uint32_t)));
  MGX_HIP_TRY(hipMemsetAsync(counts5, 0, 20, ctx->stream));

  hipLaunchKernelGGL(k_bin_keys, dim3(grid_for(rows)), dim3(kBlock), 0, ctx->stream, rows,
                     lo, hi, include_zero ? 1 : 0, keys, vals);
  hipLaunchKernelGGL(k_count_bins, dim3(grid_for(rows)), dim3(kBlock), 0, ctx->stream, rows,
                     keys, counts5);

  // Stable 3-bit radix sort: within a bin, rows stay in ascending id order;
  // key 4 (dropped rows) lands past the used prefix.
  size_t tmp_bytes = 0;
  auto err = rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys, keys_out, vals,
                                       (uint32_t *)bins->rows, rows, 0, 3, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::radix_sort_pairs(tmp, tmp_bytes, keys, keys_out, vals,
                                  (uint32_t *)bins->rows, rows, 0, 3, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;

  uint32_t h_counts[5] = {0, 0, 0, 0, 0};
  MGX_HIP_TRY(hipMemcpyAsync(h_counts, counts5, 20, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  for (int b = 0; b < 4; ++b) bins->count[b] = h_counts[b];

  // Launch geometry: rows-per-block {64,16,4,1}; grid-stride caps keep the
  // dispatch bounded (guide G11) while >> 256 CUs stay fed.
  const int64_t rows_per_block[4] = {64, 16, 4, 1};
  const int64_t cap[4] = {2048, 2048, 2048, 8192};
  for (int b = 0; b < 4; ++b) {
    int64_t need = (bins->count[b] + rows_per_block[b] - 1) / rows_per_block[b];
    bins->grid[b] = bins->count[b] ? (need < cap[b] ? need : cap[b]) : 0;
  }

  MGX_HIP_TRY(hipFree(keys));
  MGX_HIP_TRY(hipFree(keys_out));
  MGX_HIP_TRY(hipFree(vals));
  MGX_HIP_TRY(hipFree(counts5));
  return MGX_OK;
}

mgx_status mgx_build_bins(mgx_context *ctx, const uint32_t *row_ptr, int64_t rows,
                          mgx_bins *bins) {
  return mgx_build_bins_range(ctx, row_ptr, row_ptr + 1, rows, /*include_zero=*/true,
                              bins);
}

namespace {
// Binary-search the sorted cols of each row for the stripe boundary value.
__global__ void k_stripe_search(int64_t rows, const uint32_t *row_ptr, const int32_t *col,
                                int32_t boundary, uint32_t *out) {
  for (int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; r < rows;
       r += (int64_t)gridDim.x * blockDim.x) {
    uint32_t lo = row_ptr[r], hi = row_ptr[r + 1];
    while (lo < hi) {
      uint32_t mid = (lo + hi) / 2;
      if (col[mid] < boundary) lo = mid + 1;
      else hi = mid;
    }
    out[r] = lo;
  }
}
}  // namespace

mgx_status mgx_build_stripes(mgx_context *ctx, mgx_graph *g) {
  // Called while row_end is still the clamped row count (the sharded path
  // re-pads it afterwards), so this is exactly the in-CSR row count.
  const int64_t rows = g->row_end - g->row_begin;
  const int64_t V = g->n_vertices;
  int n_stripes = 1;
  const char *env = getenv("MGX_PR_STRIPES");
  if (env && atoi(env) > 0) {
    n_stripes = atoi(env);
  } else if (g->order == nullptr) {
    // auto (identity layout, e.g. sharded): keep the gathered contrib
    // stripe (f32[V/S]) under ~96 MB so it stays Infinity-Cache-resident.
    const int64_t contrib_bytes = V * 4;
    n_stripes = (int)((contrib_bytes + (96 << 20) - 1) / (96 << 20));
    if (n_stripes < 1) n_stripes = 1;
  } else {
    // Hot-first-permuted layout: the hub prefix already keeps the gather
    // working set cache-resident; stripes only add fp64-partial traffic
    // (measured RMAT-26: S=1 116.3 G edges/s vs S=3 106.0). Stay unstriped.
    n_stripes = 1;
  }
  if (n_stripes > 16) n_stripes = 16;
  g->n_stripes = n_stripes;
  if (n_stripes == 1) return MGX_OK;
  g->stripe_width = (V + n_stripes - 1) / n_stripes;

  MGX_HIP_TRY(mgx_hip_malloc(&g->stripe_ptr, (size_t)(n_stripes + 1) * rows * sizeof(uint32_t)));
  // boundaries: sp[0] = row starts, sp[S] = row ends, sp[s] = searchsorted.
  struct CopyK {
    static __global__ void shift(int64_t rows, const uint32_t *row_ptr, uint32_t *lo,
                                 uint32_t *hi) {
      for (int64_t r = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; r < rows;
           r += (int64_t)gridDim.x * blockDim.x) {
        lo[r] = row_ptr[r];
        hi[r] = row_ptr[r + 1];
      }
    }
  };
  hipLaunchKernelGGL(CopyK::shift, dim3(grid_for(rows)), dim3(kBlock), 0, ctx->stream, rows,
                     g->in_row_ptr, g->stripe_ptr, g->stripe_ptr + (size_t)n_stripes * rows);
  for (int sIdx = 1; sIdx < n_stripes; ++sIdx) {
    hipLaunchKernelGGL(k_stripe_search, dim3(grid_for(rows)), dim3(kBlock), 0, ctx->stream,
                       rows, g->in_row_ptr, g->in_col,
                       (int32_t)((int64_t)sIdx * g->stripe_width),
                       g->stripe_ptr + (size_t)sIdx * rows);
  }
  for (int sIdx = 0; sIdx < n_stripes; ++sIdx) {
    const bool include_zero = (sIdx == 0) || (sIdx == n_stripes - 1);
    MGX_TRY(mgx_build_bins_range(ctx, g->stripe_ptr + (size_t)sIdx * rows,
                                 g->stripe_ptr + (size_t)(sIdx + 1) * rows, rows,
                                 include_zero, &g->stripe_bins[sIdx]));
  }
  return MGX_OK;
}

mgx_status mgx_build_from_device_coo(mgx_context *ctx, const int32_t *d_src,
                                     const int32_t *d_dst, const float *d_w,
                                     int64_t n_vertices, int64_t n_edges, uint32_t flags,
                                     mgx_graph *g) {
  const int64_t V = n_vertices, E = n_edges;
  g->n_vertices = V;
  g->n_edges = E;
  g->flags = flags;
  g->row_begin = 0;
  g->row_end = V;
  g->in_edges = E;

  hipEvent_t ev0, ev1;
  MGX_HIP_TRY(hipEventCreate(&ev0));
  MGX_HIP_TRY(hipEventCreate(&ev1));
  MGX_HIP_TRY(hipEventRecord(ev0, ctx->stream));

  uint32_t *counts = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&counts, (V > 0 ? V : 1) * sizeof(uint32_t)));

  // out-degree (+ inv) — always needed by PageRank/Katz result paths.
  MGX_HIP_TRY(mgx_hip_malloc(&g->out_degree, (V > 0 ? V : 1) * sizeof(uint32_t)));
  MGX_HIP_TRY(hipMemsetAsync(g->out_degree, 0, V * sizeof(uint32_t), ctx->stream));
  hipLaunchKernelGGL(k_hist, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E, d_src,
                     g->out_degree);
  MGX_HIP_TRY(mgx_hip_malloc(&g->inv_outdeg, (V > 0 ? V : 1) * sizeof(float)));
  hipLaunchKernelGGL(k_inv_outdeg, dim3(grid_for(V)), dim3(kBlock), 0, ctx->stream, V,
                     g->out_degree, g->inv_outdeg);

  if ((flags & MGX_BUILD_IN_CSR) && (flags & MGX_BUILD_NO_PERM)) {
    // Identity layout (online/dynamic paths: row ids must equal scan ids).
    MGX_HIP_TRY(hipMemsetAsync(counts, 0, V * sizeof(uint32_t), ctx->stream));
    hipLaunchKernelGGL(k_hist, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E, d_dst,
                       counts);
    MGX_HIP_TRY(mgx_hip_malloc(&g->in_row_ptr, (V + 1) * sizeof(uint32_t)));
    MGX_TRY(scan_counts(ctx, counts, V, g->in_row_ptr));
    MGX_HIP_TRY(mgx_hip_malloc(&g->in_col, (E > 0 ? E : 1) * sizeof(int32_t)));
    if ((flags & MGX_BUILD_WEIGHTED) && d_w) {
      MGX_HIP_TRY(mgx_hip_malloc(&g->in_w, (E > 0 ? E : 1) * sizeof(float)));
      MGX_TRY(build_sorted_cols_w(ctx, d_src, d_dst, d_w, E, V, g->in_col, g->in_w));
    } else {
      MGX_TRY(build_sorted_cols(ctx, d_src, d_dst, nullptr, E, V, false, 0, 0, g->in_col,
                                E));
    }
    MGX_TRY(mgx_build_bins(ctx, g->in_row_ptr, V, &g->bins_in));
  } else if (flags & MGX_BUILD_IN_CSR) {
    // Hot-first vertex permutation: renumber by descending out-degree
    // (stable, ties by original id) so the most-gathered contrib entries
    // pack into the lowest addresses (L2/L3-resident under power-law skew).
    int32_t *d_perm = nullptr;
    MGX_HIP_TRY(mgx_hip_malloc(&g->order, (V > 0 ? V : 1) * sizeof(int32_t)));
    MGX_HIP_TRY(mgx_hip_malloc(&d_perm, (V > 0 ? V : 1) * sizeof(int32_t)));
    {
      uint32_t *deg_sorted = nullptr;
      int32_t *iota = nullptr;
      MGX_HIP_TRY(mgx_hip_malloc(&deg_sorted, (V > 0 ? V : 1) * sizeof(uint32_t)));
      MGX_HIP_TRY(mgx_hip_malloc(&iota, (V > 0 ? V : 1) * sizeof(int32_t)));
      hipLaunchKernelGGL(k_iota_i32g, dim3(grid_for(V)), dim3(kBlock), 0, ctx->stream, V,
                         iota);
      size_t tmp_bytes = 0;
      auto err = rocprim::radix_sort_pairs_desc(nullptr, tmp_bytes, g->out_degree,
                                                deg_sorted, iota, g->order, V, 0, 32,
                                                ctx->stream);
      if (err != hipSuccess) return MGX_ERR_HIP;
      void *tmp = nullptr;
      MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
      err = rocprim::radix_sort_pairs_desc(tmp, tmp_bytes, g->out_degree, deg_sorted, iota,
                                           g->order, V, 0, 32, ctx->stream);
      if (err != hipSuccess) return MGX_ERR_HIP;
      hipLaunchKernelGGL(k_invert_perm, dim3(grid_for(V)), dim3(kBlock), 0, ctx->stream, V,
                         g->order, d_perm);
      // out_degree / inv_outdeg move to the permuted space.
      MGX_HIP_TRY(hipMemcpyAsync(g->out_degree, deg_sorted, V * sizeof(uint32_t),
                                 hipMemcpyDeviceToDevice, ctx->stream));
      hipLaunchKernelGGL(k_inv_outdeg, dim3(grid_for(V)), dim3(kBlock), 0, ctx->stream, V,
                         g->out_degree, g->inv_outdeg);
      MGX_HIP_TRY(hipFree(deg_sorted));
      MGX_HIP_TRY(hipFree(iota));
    }
    MGX_HIP_TRY(hipMemsetAsync(counts, 0, V * sizeof(uint32_t), ctx->stream));
    hipLaunchKernelGGL(k_hist_perm, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E,
                       d_dst, d_perm, counts);
    MGX_HIP_TRY(mgx_hip_malloc(&g->in_row_ptr, (V + 1) * sizeof(uint32_t)));
    MGX_TRY(scan_counts(ctx, counts, V, g->in_row_ptr));
    MGX_HIP_TRY(mgx_hip_malloc(&g->in_col, (E > 0 ? E : 1) * sizeof(int32_t)));
    MGX_TRY(build_sorted_cols(ctx, d_src, d_dst, d_perm, E, V, false, 0, 0, g->in_col, E));
    MGX_HIP_TRY(hipFree(d_perm));
    MGX_TRY(mgx_build_bins(ctx, g->in_row_ptr, V, &g->bins_in));
    MGX_TRY(mgx_build_stripes(ctx, g));
  }

  if (flags & MGX_BUILD_OUT_CSR) {
    // out-CSR in the ORIGINAL vertex space (Brandes keys results by scan
    // ids and traverses forward edges; no hot-first perm here).
    MGX_HIP_TRY(hipMemsetAsync(counts, 0, V * sizeof(uint32_t), ctx->stream));
    hipLaunchKernelGGL(k_hist, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E, d_src,
                       counts);
    MGX_HIP_TRY(mgx_hip_malloc(&g->out_row_ptr, (V + 1) * sizeof(uint32_t)));
    MGX_TRY(scan_counts(ctx, counts, V, g->out_row_ptr));
    MGX_HIP_TRY(mgx_hip_malloc(&g->out_col, (E > 0 ? E : 1) * sizeof(int32_t)));
    // reuse the sorted-cols builder with (src,dst) swapped: rows = sources
    MGX_TRY(build_sorted_cols(ctx, d_dst, d_src, nullptr, E, V, false, 0, 0, g->out_col,
                              E));
    MGX_TRY(mgx_build_bins(ctx, g->out_row_ptr, V, &g->bins_out));
  }

  if (flags & MGX_BUILD_SYM_CSR) {
    MGX_HIP_TRY(hipMemsetAsync(counts, 0, V * sizeof(uint32_t), ctx->stream));
    hipLaunchKernelGGL(k_hist2, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E, d_src,
                       d_dst, counts);
    MGX_HIP_TRY(mgx_hip_malloc(&g->sym_row_ptr, (V + 1) * sizeof(uint32_t)));
    MGX_TRY(scan_counts(ctx, counts, V, g->sym_row_ptr));
    MGX_HIP_TRY(mgx_hip_malloc(&g->sym_col, (E > 0 ? 2 * E : 1) * sizeof(int32_t)));
    if (flags & MGX_BUILD_WEIGHTED) {
      MGX_HIP_TRY(mgx_hip_malloc(&g->sym_w, (E > 0 ? 2 * E : 1) * sizeof(float)));
    }
    MGX_TRY(build_sorted_sym(ctx, d_src, d_dst, d_w, E, V, g->sym_col, g->sym_w));
    MGX_TRY(mgx_build_bins(ctx, g->sym_row_ptr, V, &g->bins_sym));
  }

  MGX_HIP_TRY(hipFree(counts));
  MGX_HIP_TRY(hipEventRecord(ev1, ctx->stream));
  MGX_HIP_TRY(hipEventSynchronize(ev1));
  float ms = 0.f;
  MGX_HIP_TRY(hipEventElapsedTime(&ms, ev0, ev1));
  g->build_ms = ms;
  MGX_HIP_TRY(hipEventDestroy(ev0));
  MGX_HIP_TRY(hipEventDestroy(ev1));
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}

mgx_status mgx_build_sharded_in_csr(mgx_context *ctx, const int32_t *d_src,
                                    const int32_t *d_dst, int64_t n_vertices,
                                    int64_t n_edges, int64_t row_begin, int64_t row_end,
                                    mgx_graph *g) {
  const int64_t V = n_vertices, E = n_edges;
  const int64_t rows = row_end - row_begin;
  g->n_vertices = V;
  g->n_edges = E;
  g->flags = MGX_BUILD_IN_CSR;
  g->row_begin = row_begin;
  g->row_end = row_end;

  hipEvent_t ev0, ev1;
  MGX_HIP_TRY(hipEventCreate(&ev0));
  MGX_HIP_TRY(hipEventCreate(&ev1));
  MGX_HIP_TRY(hipEventRecord(ev0, ctx->stream));

  // Global out-degree (contrib denominators need every source).
  MGX_HIP_TRY(mgx_hip_malloc(&g->out_degree, V * sizeof(uint32_t)));
  MGX_HIP_TRY(hipMemsetAsync(g->out_degree, 0, V * sizeof(uint32_t), ctx->stream));
  hipLaunchKernelGGL(k_hist, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E, d_src,
                     g->out_degree);
  MGX_HIP_TRY(mgx_hip_malloc(&g->inv_outdeg, V * sizeof(float)));
  hipLaunchKernelGGL(k_inv_outdeg, dim3(grid_for(V)), dim3(kBlock), 0, ctx->stream, V,
                     g->out_degree, g->inv_outdeg);

  uint32_t *counts = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&counts, (rows > 0 ? rows : 1) * sizeof(uint32_t)));
  MGX_HIP_TRY(hipMemsetAsync(counts, 0, rows * sizeof(uint32_t), ctx->stream));
  hipLaunchKernelGGL(k_hist_ranged, dim3(grid_for(E)), dim3(kBlock), 0, ctx->stream, E,
                     d_dst, (int32_t)row_begin, (int32_t)row_end, counts);
  MGX_HIP_TRY(mgx_hip_malloc(&g->in_row_ptr, (rows + 1) * sizeof(uint32_t)));
  MGX_TRY(scan_counts(ctx, counts, rows, g->in_row_ptr));
  uint32_t local_edges = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&local_edges, g->in_row_ptr + rows, 4, hipMemcpyDeviceToHost,
                             ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  g->in_edges = local_edges;
  MGX_HIP_TRY(mgx_hip_malloc(&g->in_col, (local_edges > 0 ? local_edges : 1) * sizeof(int32_t)));
  MGX_TRY(build_sorted_cols(ctx, d_src, d_dst, nullptr, E, rows, true, (int32_t)row_begin,
                            (int32_t)row_end, g->in_col, local_edges));
  MGX_TRY(mgx_build_bins(ctx, g->in_row_ptr, rows, &g->bins_in));
  MGX_TRY(mgx_build_stripes(ctx, g));
  MGX_HIP_TRY(hipFree(counts));

  MGX_HIP_TRY(hipEventRecord(ev1, ctx->stream));
  MGX_HIP_TRY(hipEventSynchronize(ev1));
  float ms = 0.f;
  MGX_HIP_TRY(hipEventElapsedTime(&ms, ev0, ev1));
  g->build_ms = ms;
  MGX_HIP_TRY(hipEventDestroy(ev0));
  MGX_HIP_TRY(hipEventDestroy(ev1));
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}

// Host COO (int64/double) upload helpers, used by mgx_graph_from_coo.
mgx_status mgx_upload_coo(mgx_context *ctx, const int64_t *src, const int64_t *dst,
                          const double *weights, int64_t n_edges, int32_t **d_src,
                          int32_t **d_dst, float **d_w) {
  MGX_HIP_TRY(mgx_hip_malloc(d_src, (n_edges > 0 ? n_edges : 1) * sizeof(int32_t)));
  MGX_HIP_TRY(mgx_hip_malloc(d_dst, (n_edges > 0 ? n_edges : 1) * sizeof(int32_t)));
  *d_w = nullptr;
  if (weights) MGX_HIP_TRY(mgx_hip_malloc(d_w, (n_edges > 0 ? n_edges : 1) * sizeof(float)));

  // Chunked staging: int64 -> int32 converted on device.
  const int64_t chunk = 16 << 20;
  int64_t *stage64 = nullptr;
  double *stagef = nullptr;
  const int64_t this_chunk = n_edges < chunk ? n_edges : chunk;
  if (this_chunk > 0) {
    MGX_HIP_TRY(mgx_hip_malloc(&stage64, this_chunk * sizeof(int64_t)));
    if (weights) MGX_HIP_TRY(mgx_hip_malloc(&stagef, this_chunk * sizeof(double)));
  }
  for (int64_t off = 0; off < n_edges; off += chunk) {
    const int64_t n = (n_edges - off) < chunk ? (n_edges - off) : chunk;
    MGX_HIP_TRY(hipMemcpyAsync(stage64, src + off, n * sizeof(int64_t),
                               hipMemcpyHostToDevice, ctx->stream));
    hipLaunchKernelGGL(k_i64_to_i32, dim3(grid_for(n)), dim3(kBlock), 0, ctx->stream, n,
                       stage64, *d_src + off);
    MGX_HIP_TRY(hipMemcpyAsync(stage64, dst + off, n * sizeof(int64_t),
                               hipMemcpyHostToDevice, ctx->stream));
    hipLaunchKernelGGL(k_i64_to_i32, dim3(grid_for(n)), dim3(kBlock), 0, ctx->stream, n,
                       stage64, *d_dst + off);
    if (weights) {
      MGX_HIP_TRY(hipMemcpyAsync(stagef, weights + off, n * sizeof(double),
                                 hipMemcpyHostToDevice, ctx->stream));
      hipLaunchKernelGGL(k_f64_to_f32, dim3(grid_for(n)), dim3(kBlock), 0, ctx->stream, n,
                         stagef, *d_w + off);
    }
    // The next chunk reuses the staging buffer: wait for queued copies.
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }
  if (stage64) MGX_HIP_TRY(hipFree(stage64));
  if (stagef) MGX_HIP_TRY(hipFree(stagef));
  return MGX_OK;
}

namespace {
__global__ void k_i32_to_i64(int64_t n, const int32_t *in, int64_t *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int64_t)in[i];
}
}  // namespace

// Test support: device-generate an edge list and download it (int64), so
// tests can check bit-identity with the numpy/C generators.
extern "C" mgx_status mgx_gen_edges_to_host(mgx_context *ctx, int rmat, int scale,
                                            int64_t n_vertices, int64_t n_edges,
                                            uint64_t seed, double a, double b, double c,
                                            int64_t *out_src, int64_t *out_dst) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  int32_t *d_src = nullptr, *d_dst = nullptr;
  int64_t *d_wide = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_src, (n_edges > 0 ? n_edges : 1) * sizeof(int32_t)));
  MGX_HIP_TRY(mgx_hip_malloc(&d_dst, (n_edges > 0 ? n_edges : 1) * sizeof(int32_t)));
  MGX_HIP_TRY(mgx_hip_malloc(&d_wide, (n_edges > 0 ? n_edges : 1) * sizeof(int64_t)));
  mgx_status s;
  if (rmat) {
    s = mgx_gen_rmat_device(ctx, scale, n_edges, seed, a, b, c, d_src, d_dst);
  } else {
    s = mgx_gen_uniform_device(ctx, n_vertices, n_edges, seed, d_src, d_dst);
  }
  if (s == MGX_OK) {
    hipLaunchKernelGGL(k_i32_to_i64, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                       n_edges, d_src, d_wide);
    (void)hipMemcpyAsync(out_src, d_wide, n_edges * sizeof(int64_t), hipMemcpyDeviceToHost,
                         ctx->stream);
    (void)hipStreamSynchronize(ctx->stream);
    hipLaunchKernelGGL(k_i32_to_i64, dim3(grid_for(n_edges)), dim3(kBlock), 0, ctx->stream,
                       n_edges, d_dst, d_wide);
    (void)hipMemcpyAsync(out_dst, d_wide, n_edges * sizeof(int64_t), hipMemcpyDeviceToHost,
                         ctx->stream);
    (void)hipStreamSynchronize(ctx->stream);
  }
  (void)hipFree(d_src);
  (void)hipFree(d_dst);
  (void)hipFree(d_wide);
  return s;
}
