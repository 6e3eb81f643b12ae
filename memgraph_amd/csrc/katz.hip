// Katz centrality on gfx950 — replaces katz_alg::SetKatz / KatzCentralityLoop
// (reference katz.cpp:393-414, :226-255) with dense fp64 state:
//   omega_i(v) = sum_{u->v} omega_{i-1}(u)      (in-CSR gather sweep)
//   c_i = c_{i-1} + alpha^i * omega_i ; lr = c_i ; ur = c_i + alpha^{i+1}*omega_i*gamma
// Convergence replicates Converged (katz.cpp:165-215) AFTER the k-override
// at :172: stable sort all centralities descending (ties -> smaller node id,
// as the oracle documents) and require ur(v_i) - eps < lr(v_{i-1}) for every
// adjacent sorted pair. gamma = degmax/(1 - alpha^2*degmax) in IEEE
// semantics (katz.cpp:403-404) — including the divergent-series regime.

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

struct KatzArgs {
  const uint32_t *row_ptr;
  const int32_t *col;
  const int32_t *bin_rows;
  int64_t n[4];
  int64_t off[4];
  int64_t goff[4];
  int64_t grid[4];
  const double *omega_old;
  double *omega_new;
  double *centrality;
  double *lr;
  double *ur;
  double a_i;    // alpha^iteration
  double a_i1g;  // alpha^(iteration+1) * gamma
};

template <int LANES>
__device__ inline void katz_rows(const KatzArgs &A, int sec, int64_t block_in_sec) {
  constexpr int RPB = kBlock / LANES;
  const int64_t nrows = A.n[sec];
  const int32_t *rows_list = A.bin_rows + A.off[sec];
  const int sub = threadIdx.x % LANES;
  __shared__ double red[4];
  for (int64_t base = block_in_sec * RPB; base < nrows; base += A.grid[sec] * RPB) {
    const int64_t ri = base + threadIdx.x / LANES;
    double acc = 0.0;
    int32_t row = -1;
    if (ri < nrows) {
      row = rows_list[ri];
      const uint32_t s = A.row_ptr[row], e = A.row_ptr[row + 1];
      if constexpr (LANES >= 64) {
        // Same access shape as the PageRank sweep: scalar head to 16-B
        // alignment, int4 nontemporal col loads, 4 gathers in flight.
        uint32_t s_al = (s + 3u) & ~3u;
        if (s_al > e) s_al = e;
        for (uint32_t j = s + sub; j < s_al; j += LANES) acc += A.omega_old[A.col[j]];
        const uint32_t nvec = (e - s_al) / 4;
        typedef int v4i __attribute__((ext_vector_type(4)));
        const v4i *col4 = reinterpret_cast<const v4i *>(A.col + s_al);
        for (uint32_t c = sub; c < nvec; c += LANES) {
          const v4i cc = __builtin_nontemporal_load(col4 + c);
          acc += A.omega_old[cc.x];
          acc += A.omega_old[cc.y];
          acc += A.omega_old[cc.z];
          acc += A.omega_old[cc.w];
        }
        for (uint32_t j = s_al + nvec * 4 + sub; j < e; j += LANES)
          acc += A.omega_old[A.col[j]];
      } else {
        for (uint32_t j = s + sub; j < e; j += LANES)
          acc += A.omega_old[__builtin_nontemporal_load(A.col + j)];
      }
    }
    if constexpr (LANES <= 64) {
      for (int o = LANES / 2; o; o >>= 1) acc += __shfl_down(acc, o, LANES);
    } else {
      for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
      if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
      __syncthreads();
      if (threadIdx.x == 0) acc = red[0] + red[1] + red[2] + red[3];
    }
    if (sub == 0 && row >= 0 && (LANES <= 64 || threadIdx.x == 0)) {
      A.omega_new[row] = acc;
      const double c = A.centrality[row] + A.a_i * acc;
      A.centrality[row] = c;
      A.lr[row] = c;                       // katz.cpp:247
      A.ur[row] = c + A.a_i1g * acc;       // katz.cpp:248-250
    }
    if constexpr (LANES > 64) __syncthreads();
  }
}

__global__ void __launch_bounds__(kBlock) k_katz_sweep(KatzArgs A) {
  const int64_t b = blockIdx.x;
  int sec = 3;
  if (b < A.goff[1]) sec = 0;
  else if (b < A.goff[2]) sec = 1;
  else if (b < A.goff[3]) sec = 2;
  const int64_t bis = b - A.goff[sec];
  switch (sec) {
    case 0: katz_rows<4>(A, 0, bis); break;
    case 1: katz_rows<16>(A, 1, bis); break;
    case 2: katz_rows<64>(A, 2, bis); break;
    default: katz_rows<256>(A, 3, bis); break;
  }
}

__global__ void k_scatter_f64(int64_t n, const double *in, const int32_t *order,
                              double *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[order ? order[i] : i] = in[i];
}

__global__ void k_fill_f64(int64_t n, double v, double *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = v;
}

// Stage the convergence-sort input in ORIGINAL vertex order so the stable
// descending radix sort breaks centrality ties by original id — the
// oracle's (documented) tie rule — while values stay permuted indices for
// the lr/ur lookups.
__global__ void k_sort_stage(int64_t n, const double *cent, const int32_t *order,
                             double *keys, uint32_t *vals) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t old = order ? order[i] : i;
    keys[old] = cent[i];
    vals[old] = (uint32_t)i;
  }
}

__global__ void k_max_u32(int64_t n, const uint32_t *x, uint32_t *out) {
  uint32_t m = 0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    m = max(m, x[i]);
  __shared__ uint32_t red[kBlock / 64];
  for (int o = 32; o; o >>= 1) m = max(m, __shfl_down(m, o, 64));
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = m;
  __syncthreads();
  if (threadIdx.x == 0) {
    uint32_t s = red[0];
    for (int i = 1; i < kBlock / 64; ++i) s = max(s, red[i]);
    atomicMax(out, s);
  }
}

// Adjacent-pair convergence test over the descending-sorted order
// (katz.cpp:206-213): violation if ur(order[i]) - eps >= lr(order[i-1]).
__global__ void k_katz_check(int64_t n, const uint32_t *order, const double *lr,
                             const double *ur, double eps, uint32_t *violated) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x + 1; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (ur[order[i]] - eps >= lr[order[i - 1]]) atomicOr(violated, 1u);
  }
}

}  // namespace

mgx_status mgx_katz_impl(mgx_context *ctx, mgx_graph *g, double alpha, double epsilon,
                         double *out_centrality, int64_t *iterations) {
  if (!(g->flags & MGX_BUILD_IN_CSR)) {
    mgx_set_error("katz needs a graph built with MGX_BUILD_IN_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  const int64_t V = g->n_vertices;
  if (iterations) *iterations = 0;
  if (V == 0) return MGX_OK;
  if (out_centrality) {
    for (int64_t v = 0; v < V; ++v) out_centrality[v] = 0.0;
  }
  // SetKatz early-outs on an edgeless graph (katz.cpp:398-400).
  if (g->n_edges == 0) return MGX_OK;

  // MaxDegree over OUT-degrees (katz.cpp:137-148 / mg_graph.hpp:96-109).
  uint32_t *d_max = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&d_max, 4));
  MGX_HIP_TRY(hipMemsetAsync(d_max, 0, 4, ctx->stream));
  hipLaunchKernelGGL(k_max_u32, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, g->out_degree, d_max);
  uint32_t deg_max = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&deg_max, d_max, 4, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_TRY(ctx->free_async(d_max));
  const double gamma = (double)deg_max / (1.0 - alpha * alpha * (double)deg_max);

  double *omega[2] = {nullptr, nullptr}, *cent = nullptr, *lr = nullptr, *ur = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&omega[0], V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&omega[1], V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&cent, V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&lr, V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&ur, V * sizeof(double)));
  hipLaunchKernelGGL(k_fill_f64, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, 1.0, omega[0]);  // omega_0 = 1 (katz.cpp:41-44)
  MGX_HIP_TRY(hipMemsetAsync(cent, 0, V * sizeof(double), ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(lr, 0, V * sizeof(double), ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(ur, 0, V * sizeof(double), ctx->stream));

  // Sort buffers for the convergence test.
  double *keys_in = nullptr, *keys_out = nullptr;
  uint32_t *vals_in = nullptr, *vals_out = nullptr, *d_flag = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&keys_in, V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&keys_out, V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&vals_in, V * sizeof(uint32_t)));
  MGX_TRY(ctx->alloc_async((void **)&vals_out, V * sizeof(uint32_t)));
  MGX_TRY(ctx->alloc_async((void **)&d_flag, 4));

  KatzArgs A;
  A.row_ptr = g->in_row_ptr;
  A.col = g->in_col;
  A.bin_rows = g->bins_in.rows;
  int64_t off = 0, goff = 0;
  for (int b = 0; b < 4; ++b) {
    A.n[b] = g->bins_in.count[b];
    A.off[b] = off;
    off += A.n[b];
    A.goff[b] = goff;
    A.grid[b] = g->bins_in.grid[b];
    goff += A.grid[b];
  }
  A.centrality = cent;
  A.lr = lr;
  A.ur = ur;

  int cur = 0;
  int64_t iter = 0;
  mgx_status status = MGX_OK;
  while (true) {
    ++iter;
    A.omega_old = omega[cur];
    A.omega_new = omega[1 - cur];
    A.a_i = pow(alpha, (double)iter);
    A.a_i1g = pow(alpha, (double)(iter + 1)) * gamma;
    hipLaunchKernelGGL(k_katz_sweep, dim3((uint32_t)goff), dim3(kBlock), 0, ctx->stream, A);
    cur = 1 - cur;

    // Stable descending sort of (centrality, original id): the stage kernel
    // places entries in original-id order, so stability gives the oracle's
    // tie rule; values are permuted indices for the lr/ur lookups.
    hipLaunchKernelGGL(k_sort_stage, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0,
                       ctx->stream, V, cent, g->order, keys_in, vals_in);
    size_t tmp_bytes = 0;
    auto err = rocprim::radix_sort_pairs_desc(nullptr, tmp_bytes, keys_in, keys_out,
                                              vals_in, vals_out, V, 0, 64, ctx->stream);
    if (err != hipSuccess) { status = MGX_ERR_HIP; break; }
    void *tmp = nullptr;
    status = ctx->reserve(tmp_bytes, &tmp);
    if (status != MGX_OK) break;
    err = rocprim::radix_sort_pairs_desc(tmp, tmp_bytes, keys_in, keys_out, vals_in,
                                         vals_out, V, 0, 64, ctx->stream);
    if (err != hipSuccess) { status = MGX_ERR_HIP; break; }

    MGX_HIP_TRY(hipMemsetAsync(d_flag, 0, 4, ctx->stream));
    hipLaunchKernelGGL(k_katz_check, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0,
                       ctx->stream, V, vals_out, lr, ur, epsilon, d_flag);
    uint32_t violated = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&violated, d_flag, 4, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (!violated) break;
    if (iter > 1000000) {  // safety net, not in reference
      mgx_set_error("katz did not converge in 1e6 iterations");
      status = MGX_ERR_INVALID_ARGUMENT;
      break;
    }
  }

  if (status == MGX_OK && out_centrality) {
    // in-CSR lives in the hot-first permuted space: scatter back to
    // original ids on device, then one contiguous D2H.
    double *scat = keys_in;  // reuse
    hipLaunchKernelGGL(k_scatter_f64, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0,
                       ctx->stream, V, cent, g->order, scat);
    MGX_HIP_TRY(hipMemcpyAsync(out_centrality, scat, V * sizeof(double),
                               hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }
  if (iterations) *iterations = iter;

  MGX_TRY(ctx->free_async(omega[0]));
  MGX_TRY(ctx->free_async(omega[1]));
  MGX_TRY(ctx->free_async(cent));
  MGX_TRY(ctx->free_async(lr));
  MGX_TRY(ctx->free_async(ur));
  MGX_TRY(ctx->free_async(keys_in));
  MGX_TRY(ctx->free_async(keys_out));
  MGX_TRY(ctx->free_async(vals_in));
  MGX_TRY(ctx->free_async(vals_out));
  MGX_TRY(ctx->free_async(d_flag));
  MGX_HIP_TRY(hipGetLastError());
  return status;
}
