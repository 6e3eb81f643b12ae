// Betweenness centrality on gfx950 — replaces
// betweenness_centrality_alg::BetweennessCentrality (reference
// betweenness_centrality_module/algorithm/betweenness_centrality.cpp:71-147):
// exact Brandes, O(V*E), all-sources.
//
// GPU shape: SOURCE-BATCHED level-synchronous Brandes. A batch of B sources
// advances together; per (source, vertex) state lives in [B][V] arrays
// (dist i32, sigma u64 with the reference's wrap semantics, dep f64).
// Forward: frontier-by-level sigma accumulation (atomicCAS on dist +
// atomicAdd on sigma — the multiset of contributions equals the
// reference's predecessor-list walk). Backward: pull-form dependency
// accumulation over the SAME adjacency (dep[u] += sigma[u]/sigma[w] *
// (1+dep[w]) for successors w one level deeper) — each dep[u] is written
// by one thread in adjacency order, so per-source dependencies are
// deterministic; only the final cross-source bc accumulation uses fp64
// atomics (the reference's own thread pool does the same,
// betweenness_centrality.cpp:103).
//
// directed: traverses the out-CSR; undirected: the symmetric CSR (multi-
// edge multiplicity affects path counts exactly as the reference's
// duplicated adjacency entries do), dependencies halved
// (betweenness_centrality.cpp:102).

#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;

inline int64_t grid_for(int64_t work, int64_t cap = 8192) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

__global__ void k_bc_init_batch(int64_t n, int32_t *dist, unsigned long long *sigma,
                                double *dep) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    dist[i] = -1;
    sigma[i] = 0;
    dep[i] = 0.0;
  }
}

__global__ void k_bc_seed(int64_t batch, int64_t s0, int64_t V, int32_t *dist,
                          unsigned long long *sigma) {
  for (int64_t b = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; b < batch;
       b += (int64_t)gridDim.x * blockDim.x) {
    const int64_t s = s0 + b;
    dist[b * V + s] = 0;
    sigma[b * V + s] = 1;
  }
}

// One forward level for the whole batch: expand vertices at `level`.
__global__ void k_bc_forward(int64_t batch, int64_t V, const uint32_t *row_ptr,
                             const int32_t *col, int32_t *dist,
                             unsigned long long *sigma, int32_t level,
                             uint32_t *changed) {
  bool any = false;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < batch * V;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t b = i / V;
    const int64_t u = i % V;
    if (dist[i] != level) continue;
    const unsigned long long su = sigma[i];
    const int64_t base = b * V;
    for (uint32_t j = row_ptr[u]; j < row_ptr[u + 1]; ++j) {
      const int32_t w = col[j];
      int32_t old = atomicCAS(&dist[base + w], -1, level + 1);
      if (old == -1 || old == level + 1) {
        atomicAdd(&sigma[base + w], su);
        any = true;
      }
    }
  }
  if (any) atomicOr(changed, 1u);
}

// One backward level: pull dependencies into vertices at `level` from their
// successors at level+1.
__global__ void k_bc_backward(int64_t batch, int64_t V, const uint32_t *row_ptr,
                              const int32_t *col, const int32_t *dist,
                              const unsigned long long *sigma, double *dep,
                              int32_t level) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < batch * V;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t b = i / V;
    const int64_t u = i % V;
    if (dist[i] != level) continue;
    const int64_t base = b * V;
    double acc = 0.0;
    const double su = (double)sigma[i];
    for (uint32_t j = row_ptr[u]; j < row_ptr[u + 1]; ++j) {
      const int32_t w = col[j];
      if (dist[base + w] == level + 1) {
        acc += (su / (double)sigma[base + w]) * (1.0 + dep[base + w]);
      }
    }
    dep[i] += acc;
  }
}

__global__ void k_bc_accumulate(int64_t batch, int64_t s0, int64_t V, const int32_t *dist,
                                const double *dep, int directed, double *bc) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < batch * V;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int64_t b = i / V;
    const int64_t v = i % V;
    if (v == s0 + b || dist[i] < 0) continue;
    const double d = dep[i];
    if (d != 0.0) atomicAdd(&bc[v], directed ? d : d / 2.0);
  }
}

__global__ void k_bc_normalize(int64_t V, double constant, double *bc) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < V;
       i += (int64_t)gridDim.x * blockDim.x)
    bc[i] *= constant;
}

}  // namespace

mgx_status mgx_betweenness_impl(mgx_context *ctx, mgx_graph *g, int directed, int normalize,
                                double *out_bc) {
  const uint32_t *row_ptr = directed ? g->out_row_ptr : g->sym_row_ptr;
  const int32_t *col = directed ? g->out_col : g->sym_col;
  if (!row_ptr) {
    mgx_set_error("betweenness needs MGX_BUILD_%s_CSR", directed ? "OUT" : "SYM");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  const int64_t V = g->n_vertices;
  if (V == 0) return MGX_OK;

  // Batch size: bounded by state memory (20 B per (source, vertex)).
  int64_t batch = 64;
  const char *env = getenv("MGX_BC_BATCH");
  if (env && atoi(env) > 0) batch = atoi(env);
  while (batch > 1 && batch * V * 20 > (4ll << 30)) batch /= 2;

  int32_t *dist = nullptr;
  unsigned long long *sigma = nullptr;
  double *dep = nullptr, *bc = nullptr;
  uint32_t *d_changed = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&dist, batch * V * sizeof(int32_t)));
  MGX_TRY(ctx->alloc_async((void **)&sigma, batch * V * sizeof(unsigned long long)));
  MGX_TRY(ctx->alloc_async((void **)&dep, batch * V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&bc, V * sizeof(double)));
  MGX_TRY(ctx->alloc_async((void **)&d_changed, 4));
  MGX_HIP_TRY(hipMemsetAsync(bc, 0, V * sizeof(double), ctx->stream));

  for (int64_t s0 = 0; s0 < V; s0 += batch) {
    const int64_t nb = (s0 + batch <= V) ? batch : (V - s0);
    hipLaunchKernelGGL(k_bc_init_batch, dim3((uint32_t)grid_for(nb * V)), dim3(kBlock), 0,
                       ctx->stream, nb * V, dist, sigma, dep);
    hipLaunchKernelGGL(k_bc_seed, dim3((uint32_t)grid_for(nb)), dim3(kBlock), 0,
                       ctx->stream, nb, s0, V, dist, sigma);
    // forward BFS levels
    int32_t level = 0;
    while (true) {
      MGX_HIP_TRY(hipMemsetAsync(d_changed, 0, 4, ctx->stream));
      hipLaunchKernelGGL(k_bc_forward, dim3((uint32_t)grid_for(nb * V)), dim3(kBlock), 0,
                         ctx->stream, nb, V, row_ptr, col, dist, sigma, level, d_changed);
      uint32_t h_changed = 0;
      MGX_HIP_TRY(hipMemcpyAsync(&h_changed, d_changed, 4, hipMemcpyDeviceToHost,
                                 ctx->stream));
      MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
      if (!h_changed) break;
      ++level;
      if (level > V) {
        mgx_set_error("betweenness forward BFS exceeded V levels");
        return MGX_ERR_HIP;
      }
    }
    // backward from deepest reached level down to 1 (dep of the source is
    // never emitted; level 0 pulls are still needed for dep of sources'
    // predecessors-of-successors — matching the reference's stack walk).
    for (int32_t l = level - 1; l >= 0; --l) {
      hipLaunchKernelGGL(k_bc_backward, dim3((uint32_t)grid_for(nb * V)), dim3(kBlock), 0,
                         ctx->stream, nb, V, row_ptr, col, dist, sigma, dep, l);
    }
    hipLaunchKernelGGL(k_bc_accumulate, dim3((uint32_t)grid_for(nb * V)), dim3(kBlock), 0,
                       ctx->stream, nb, s0, V, dist, dep, directed, bc);
  }

  if (normalize) {
    // betweenness_centrality.cpp:139-144
    const double pairs = (double)((V - 1) * (V - 2));
    const double numerator = directed ? 1.0 : 2.0;
    const double constant = V > 2 ? numerator / pairs : 1.0;
    hipLaunchKernelGGL(k_bc_normalize, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0,
                       ctx->stream, V, constant, bc);
  }
  if (out_bc) {
    MGX_HIP_TRY(hipMemcpyAsync(out_bc, bc, V * sizeof(double), hipMemcpyDeviceToHost,
                               ctx->stream));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_TRY(ctx->free_async(dist));
  MGX_TRY(ctx->free_async(sigma));
  MGX_TRY(ctx->free_async(dep));
  MGX_TRY(ctx->free_async(bc));
  MGX_TRY(ctx->free_async(d_changed));
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}
