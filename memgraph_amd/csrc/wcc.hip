// Weakly connected components on gfx950 — replaces Weak
// (reference connectivity_module.cpp:41-87).
//
// Min-label propagation over the symmetric CSR with pointer-jumping
// compression; at the fixpoint label[v] = min vertex id of v's component.
// Component ids are then renumbered by ascending representative id, which
// is EXACTLY the reference's BFS discovery order (the reference roots its
// BFS at the first unvisited vertex in scan order, i.e. each component is
// discovered at its min member — DESIGN.md), so emitted ids are bit-exact.

#include <cstring>

#include <rocprim/rocprim.hpp>

#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

__global__ void k_iota(int64_t n, int32_t *label) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    label[i] = (int32_t)i;
}

struct WccArgs {
  const uint32_t *row_ptr;
  const int32_t *col;
  const int32_t *bin_rows;
  int64_t n[4];
  int64_t off[4];
  int64_t goff[4];
  int64_t grid[4];
  int32_t *label;       // in-place (async min-propagation is monotone-safe)
  uint32_t *changed;
};

template <int LANES>
__device__ inline bool wcc_rows(const WccArgs &A, int sec, int64_t block_in_sec) {
  constexpr int RPB = kBlock / LANES;
  const int64_t nrows = A.n[sec];
  const int32_t *rows_list = A.bin_rows + A.off[sec];
  const int sub = threadIdx.x % LANES;
  bool changed = false;
  __shared__ int32_t red[4];
  for (int64_t base = block_in_sec * RPB; base < nrows; base += A.grid[sec] * RPB) {
    const int64_t ri = base + threadIdx.x / LANES;
    int32_t m = INT32_MAX;
    int32_t row = -1;
    if (ri < nrows) {
      row = rows_list[ri];
      const uint32_t s = A.row_ptr[row], e = A.row_ptr[row + 1];
      if constexpr (LANES >= 64) {
        uint32_t s_al = (s + 3u) & ~3u;
        if (s_al > e) s_al = e;
        for (uint32_t j = s + sub; j < s_al; j += LANES) m = min(m, A.label[A.col[j]]);
        const uint32_t nvec = (e - s_al) / 4;
        typedef int v4i __attribute__((ext_vector_type(4)));
        const v4i *col4 = reinterpret_cast<const v4i *>(A.col + s_al);
        for (uint32_t c = sub; c < nvec; c += LANES) {
          const v4i cc = __builtin_nontemporal_load(col4 + c);
          m = min(m, A.label[cc.x]);
          m = min(m, A.label[cc.y]);
          m = min(m, A.label[cc.z]);
          m = min(m, A.label[cc.w]);
        }
        for (uint32_t j = s_al + nvec * 4 + sub; j < e; j += LANES)
          m = min(m, A.label[A.col[j]]);
      } else {
        for (uint32_t j = s + sub; j < e; j += LANES)
          m = min(m, A.label[__builtin_nontemporal_load(A.col + j)]);
      }
    }
    if constexpr (LANES <= 64) {
      for (int o = LANES / 2; o; o >>= 1) m = min(m, __shfl_down(m, o, LANES));
    } else {
      for (int o = 32; o; o >>= 1) m = min(m, __shfl_down(m, o, 64));
      if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m;
      __syncthreads();
      if (threadIdx.x == 0) m = min(min(red[0], red[1]), min(red[2], red[3]));
    }
    if (sub == 0 && row >= 0 && (LANES <= 64 || threadIdx.x == 0)) {
      if (m < A.label[row]) {
        A.label[row] = m;
        changed = true;
      }
    }
    if constexpr (LANES > 64) __syncthreads();
  }
  return changed;
}

__global__ void __launch_bounds__(kBlock) k_wcc_sweep(WccArgs A) {
  const int64_t b = blockIdx.x;
  int sec = 3;
  if (b < A.goff[1]) sec = 0;
  else if (b < A.goff[2]) sec = 1;
  else if (b < A.goff[3]) sec = 2;
  const int64_t bis = b - A.goff[sec];
  bool ch;
  switch (sec) {
    case 0: ch = wcc_rows<4>(A, 0, bis); break;
    case 1: ch = wcc_rows<16>(A, 1, bis); break;
    case 2: ch = wcc_rows<64>(A, 2, bis); break;
    default: ch = wcc_rows<256>(A, 3, bis); break;
  }
  if (ch) atomicOr(A.changed, 1u);
}

// Pointer jumping: label[v] <- label[label[v]] until per-pass fixpoint
// (monotone decreasing; races only skip a shortcut, never break it).
__global__ void k_wcc_jump(int64_t n, int32_t *label, uint32_t *changed) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t l = label[i];
    const int32_t ll = label[l];
    if (ll < l) {
      label[i] = ll;
      atomicOr(changed, 1u);
    }
  }
}

// Renumber by ascending representative id: flag roots, exclusive-scan,
// comp[v] = scan[label[v]].
__global__ void k_wcc_flag_roots(int64_t n, const int32_t *label, uint32_t *flag) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    flag[i] = (label[i] == (int32_t)i) ? 1u : 0u;
}

__global__ void k_wcc_emit(int64_t n, const int32_t *label, const uint32_t *scan,
                           int64_t *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int64_t)scan[label[i]];
}

}  // namespace

mgx_status mgx_wcc_impl(mgx_context *ctx, mgx_graph *g, int64_t *out_component,
                        int64_t *n_components) {
  if (!(g->flags & MGX_BUILD_SYM_CSR)) {
    mgx_set_error("wcc needs a graph built with MGX_BUILD_SYM_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  const int64_t V = g->n_vertices;
  if (V == 0) {
    if (n_components) *n_components = 0;
    return MGX_OK;
  }

  int32_t *label = nullptr;
  uint32_t *d_changed = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&label, V * sizeof(int32_t)));
  MGX_TRY(ctx->alloc_async((void **)&d_changed, sizeof(uint32_t)));
  hipLaunchKernelGGL(k_iota, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream, V,
                     label);

  WccArgs A;
  A.row_ptr = g->sym_row_ptr;
  A.col = g->sym_col;
  A.bin_rows = g->bins_sym.rows;
  int64_t off = 0, goff = 0;
  for (int b = 0; b < 4; ++b) {
    A.n[b] = g->bins_sym.count[b];
    A.off[b] = off;
    off += A.n[b];
    A.goff[b] = goff;
    A.grid[b] = g->bins_sym.grid[b];
    goff += A.grid[b];
  }
  A.label = label;
  A.changed = d_changed;

  uint32_t h_changed = 1;
  while (h_changed) {
    MGX_HIP_TRY(hipMemsetAsync(d_changed, 0, 4, ctx->stream));
    if (goff > 0)
      hipLaunchKernelGGL(k_wcc_sweep, dim3((uint32_t)goff), dim3(kBlock), 0, ctx->stream, A);
    hipLaunchKernelGGL(k_wcc_jump, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                       V, label, d_changed);
    MGX_HIP_TRY(hipMemcpyAsync(&h_changed, d_changed, 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }

  // Renumber + count.
  uint32_t *flag = nullptr, *scan = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&flag, V * sizeof(uint32_t)));
  MGX_TRY(ctx->alloc_async((void **)&scan, (V + 1) * sizeof(uint32_t)));
  hipLaunchKernelGGL(k_wcc_flag_roots, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0,
                     ctx->stream, V, label, flag);
  size_t tmp_bytes = 0;
  auto err = rocprim::exclusive_scan(nullptr, tmp_bytes, flag, scan, 0u, V,
                                     rocprim::plus<uint32_t>(), ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::exclusive_scan(tmp, tmp_bytes, flag, scan, 0u, V,
                                rocprim::plus<uint32_t>(), ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;

  int64_t *d_out = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&d_out, V * sizeof(int64_t)));
  hipLaunchKernelGGL(k_wcc_emit, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, label, scan, d_out);
  if (out_component) {
    MGX_HIP_TRY(hipMemcpyAsync(out_component, d_out, V * sizeof(int64_t),
                               hipMemcpyDeviceToHost, ctx->stream));
  }
  if (n_components) {
    uint32_t last_scan = 0, last_flag = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&last_scan, scan + V - 1, 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(&last_flag, flag + V - 1, 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    *n_components = (int64_t)last_scan + last_flag;
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_TRY(ctx->free_async(label));
  MGX_TRY(ctx->free_async(d_changed));
  MGX_TRY(ctx->free_async(flag));
  MGX_TRY(ctx->free_async(scan));
  MGX_TRY(ctx->free_async(d_out));
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}
