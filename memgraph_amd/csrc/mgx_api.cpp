// libmgx_analytics host API: context/graph lifecycle, error reporting.
// Compiled by hipcc as HIP host code (no kernels here).

#include <cstdarg>
#include <cstdio>
#include <cstring>

#include "mgx_internal.h"

namespace {
thread_local char g_error[1024] = {0};
}

void mgx_set_error(const char *fmt, ...) {
  va_list ap;
  va_start(ap, fmt);
  vsnprintf(g_error, sizeof(g_error), fmt, ap);
  va_end(ap);
}

extern "C" const char *mgx_last_error(void) { return g_error; }

extern "C" const char *mgx_status_string(mgx_status s) {
  switch (s) {
    case MGX_OK: return "ok";
    case MGX_ERR_NO_DEVICE: return "no HIP device";
    case MGX_ERR_HIP: return "HIP error";
    case MGX_ERR_INVALID_ARGUMENT: return "invalid argument";
    case MGX_ERR_TOO_LARGE: return "graph too large (needs < 2^31 vertices/edges)";
    case MGX_ERR_OUT_OF_MEMORY: return "out of device memory";
    case MGX_ERR_NCCL: return "RCCL error";
    case MGX_ERR_NOT_SUPPORTED: return "not supported";
    default: return "unknown";
  }
}

extern "C" int mgx_device_count(void) {
  int n = 0;
  if (hipGetDeviceCount(&n) != hipSuccess) return 0;
  return n;
}

namespace {
// MGX_NO_ASYNC_POOL=1 bypasses the caching free list (one hipMalloc/hipFree
// per request) — kept for allocator-bug bisection; the env name predates the
// switch from hipMallocAsync to the in-house cache.
bool no_alloc_cache() {
  static const bool v = [] {
    const char *e = getenv("MGX_NO_ASYNC_POOL");
    return e && atoi(e) != 0;
  }();
  return v;
}

// Bucket rounding: small requests to 512 B, >=2 MB to 2 MB granularity.
// Louvain/graph-build reissue identical sizes level over level and call
// over call, so rounded-exact reuse covers the hot pattern.
size_t round_bucket(size_t bytes) {
  if (bytes == 0) bytes = 1;
  const size_t kSmall = 512, kBig = 2ull << 20;
  if (bytes < kBig) return (bytes + kSmall - 1) / kSmall * kSmall;
  return (bytes + kBig - 1) / kBig * kBig;
}
}  // namespace

mgx_status mgx_context::alloc_async(void **p, size_t bytes) {
  if (no_alloc_cache()) {
    MGX_HIP_TRY(mgx_hip_malloc(p, bytes ? bytes : 1));
    return MGX_OK;
  }
  const size_t want = round_bucket(bytes);
  {
    std::lock_guard<std::mutex> lk(cache_mu);
    auto it = cache_free.lower_bound(want);
    // Accept a cached block up to 25% (+1 bucket) larger to limit waste.
    if (it != cache_free.end() && it->first <= want + want / 4 + (2ull << 20)) {
      *p = it->second;
      cache_live[*p] = it->first;
      cache_free_bytes -= it->first;
      cache_free.erase(it);
      return MGX_OK;
    }
  }
  ensure_margin(want);
  hipError_t e = mgx_hip_malloc(p, want);
  if (e == hipErrorOutOfMemory) {
    (void)hipGetLastError();
    cache_trim();  // return every cached free block, then retry once
    e = mgx_hip_malloc(p, want);
  }
  if (e != hipSuccess) {
    (void)hipGetLastError();
    mgx_set_error("mgx_hip_malloc(%zu) failed: %s", want, hipGetErrorString(e));
    return e == hipErrorOutOfMemory ? MGX_ERR_OUT_OF_MEMORY : MGX_ERR_HIP;
  }
  std::lock_guard<std::mutex> lk(cache_mu);
  cache_live[*p] = want;
  return MGX_OK;
}

mgx_status mgx_context::free_async(void *p) {
  if (!p) return MGX_OK;
  if (no_alloc_cache()) {
    MGX_HIP_TRY(hipFree(p));
    return MGX_OK;
  }
  std::lock_guard<std::mutex> lk(cache_mu);
  auto it = cache_live.find(p);
  if (it == cache_live.end()) {
    // Not cache-owned (pre-cache allocation): plain free.
    MGX_HIP_TRY(hipFree(p));
    return MGX_OK;
  }
  cache_free.insert({it->second, p});
  cache_free_bytes += it->second;
  cache_live.erase(it);
  return MGX_OK;
}

void mgx_context::ensure_margin(size_t want) {
  // Near device-memory exhaustion the ROCm 7.0.x runtime can hand out
  // bogus mappings instead of failing (overlapping VA / read-only pages —
  // measured at RMAT-26 Louvain, see profiles/r02_summary.md). Keep a
  // safety margin: release every cached free block before a malloc that
  // would leave less than ~8 GB free.
  if (cache_free_bytes == 0) return;
  size_t free_b = 0, total_b = 0;
  if (hipMemGetInfo(&free_b, &total_b) == hipSuccess && free_b < want + (8ull << 30))
    cache_trim();
}

void mgx_context::cache_trim() {
  // The blocks on the free list may still be referenced by queued stream
  // work; drain before returning their pages.
  (void)hipStreamSynchronize(stream);
  std::lock_guard<std::mutex> lk(cache_mu);
  for (auto &e : cache_free) (void)hipFree(e.second);
  cache_free.clear();
  cache_free_bytes = 0;
}

mgx_status mgx_context::reserve(size_t bytes, void **out) {
  if (bytes > workspace_bytes) {
    // Drain queued users of the old workspace before recycling it; growth
    // is rare so the sync costs nothing.
    if (workspace) {
      MGX_HIP_TRY(hipStreamSynchronize(stream));
      MGX_TRY(free_async(workspace));
    }
    workspace = nullptr;
    workspace_bytes = 0;
    size_t want = bytes + bytes / 2;
    MGX_TRY(alloc_async(&workspace, want));
    workspace_bytes = want;
  }
  *out = workspace;
  return MGX_OK;
}

extern "C" mgx_status mgx_init(int device, mgx_context **out) {
  int n = mgx_device_count();
  if (n <= 0) {
    mgx_set_error("no HIP device visible (hipGetDeviceCount=0) — "
                  "mgx_analytics has no CPU fallback by design");
    return MGX_ERR_NO_DEVICE;
  }
  if (device < 0 || device >= n) {
    mgx_set_error("device %d out of range (%d visible)", device, n);
    return MGX_ERR_INVALID_ARGUMENT;
  }
  MGX_HIP_TRY(hipSetDevice(device));
  auto *ctx = new mgx_context();
  ctx->device = device;
  if (hipStreamCreate(&ctx->stream) != hipSuccess) {
    delete ctx;
    mgx_set_error("hipStreamCreate failed");
    return MGX_ERR_HIP;
  }
  *out = ctx;
  return MGX_OK;
}

extern "C" mgx_status mgx_destroy(mgx_context *ctx) {
  if (!ctx) return MGX_OK;
  if (ctx->comm) (void)mgx_comm_destroy(ctx);
  for (int i = 0; i < 2; ++i) {
    if (ctx->louv_col[i]) (void)hipFree(ctx->louv_col[i]);
    if (ctx->louv_w[i]) (void)hipFree(ctx->louv_w[i]);
  }
  if (ctx->workspace) (void)ctx->free_async(ctx->workspace);
  ctx->cache_trim();
  for (auto &e : ctx->cache_live) (void)hipFree(e.first);  // leaked by callers
  if (ctx->stream) (void)hipStreamDestroy(ctx->stream);
  delete ctx;
  return MGX_OK;
}

extern "C" mgx_status mgx_sync(mgx_context *ctx) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  return MGX_OK;
}

// graph_build.hip
mgx_status mgx_upload_coo(mgx_context *ctx, const int64_t *src, const int64_t *dst,
                          const double *weights, int64_t n_edges, int32_t **d_src,
                          int32_t **d_dst, float **d_w);

namespace {

mgx_status check_sizes(int64_t n_vertices, int64_t n_edges) {
  if (n_vertices < 0 || n_edges < 0) {
    mgx_set_error("negative sizes");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  if (n_vertices >= (1ll << 31) || n_edges >= (1ll << 31)) {
    mgx_set_error("V=%lld E=%lld exceeds int32 CSR indices", (long long)n_vertices,
                  (long long)n_edges);
    return MGX_ERR_TOO_LARGE;
  }
  return MGX_OK;
}

}  // namespace

extern "C" mgx_status mgx_graph_from_coo(mgx_context *ctx, const int64_t *src,
                                         const int64_t *dst, const double *weights,
                                         int64_t n_vertices, int64_t n_edges, uint32_t flags,
                                         mgx_graph **out) {
  MGX_TRY(check_sizes(n_vertices, n_edges));
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  // Range check on host (module scan already produces dense ids, but the
  // ABI promises validation).
  for (int64_t e = 0; e < n_edges; ++e) {
    if (src[e] < 0 || src[e] >= n_vertices || dst[e] < 0 || dst[e] >= n_vertices) {
      mgx_set_error("edge %lld (%lld->%lld) out of range [0,%lld)", (long long)e,
                    (long long)src[e], (long long)dst[e], (long long)n_vertices);
      return MGX_ERR_INVALID_ARGUMENT;
    }
  }
  int32_t *d_src = nullptr, *d_dst = nullptr;
  float *d_w = nullptr;
  MGX_TRY(mgx_upload_coo(ctx, src, dst, weights, n_edges, &d_src, &d_dst, &d_w));
  auto *g = new mgx_graph();
  mgx_status s = mgx_build_from_device_coo(ctx, d_src, d_dst, d_w, n_vertices, n_edges,
                                           flags, g);
  (void)hipFree(d_src);
  (void)hipFree(d_dst);
  if (d_w) (void)hipFree(d_w);
  if (s != MGX_OK) {
    (void)mgx_graph_destroy(ctx, g);
    return s;
  }
  *out = g;
  return MGX_OK;
}

namespace {

mgx_status build_generated(mgx_context *ctx, int64_t n_vertices, int64_t n_edges,
                           uint32_t flags, uint64_t weight_seed, bool rmat, int scale,
                           uint64_t seed, double a, double b, double c, mgx_graph **out) {
  MGX_TRY(check_sizes(n_vertices, n_edges));
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  int32_t *d_src = nullptr, *d_dst = nullptr;
  float *d_w = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&d_src, n_edges * sizeof(int32_t)));
  MGX_TRY(ctx->alloc_async((void **)&d_dst, n_edges * sizeof(int32_t)));
  mgx_status s;
  if (rmat) {
    s = mgx_gen_rmat_device(ctx, scale, n_edges, seed, a, b, c, d_src, d_dst);
  } else {
    s = mgx_gen_uniform_device(ctx, n_vertices, n_edges, seed, d_src, d_dst);
  }
  if (s == MGX_OK && (flags & MGX_BUILD_WEIGHTED)) {
    MGX_TRY(ctx->alloc_async((void **)&d_w, n_edges * sizeof(float)));
    s = mgx_gen_weights_device(ctx, n_edges, weight_seed, d_w);
  }
  mgx_graph *g = nullptr;
  if (s == MGX_OK) {
    g = new mgx_graph();
    s = mgx_build_from_device_coo(ctx, d_src, d_dst, d_w, n_vertices, n_edges, flags, g);
  }
  (void)ctx->free_async(d_src);
  (void)ctx->free_async(d_dst);
  if (d_w) (void)ctx->free_async(d_w);
  if (s != MGX_OK) {
    if (g) (void)mgx_graph_destroy(ctx, g);
    return s;
  }
  *out = g;
  return MGX_OK;
}

}  // namespace

extern "C" mgx_status mgx_graph_rmat(mgx_context *ctx, int scale, int64_t n_edges,
                                     uint64_t seed, double a, double b, double c,
                                     uint32_t flags, uint64_t weight_seed, mgx_graph **out) {
  if (scale < 0 || scale > 30) {
    mgx_set_error("rmat scale %d out of range", scale);
    return MGX_ERR_INVALID_ARGUMENT;
  }
  return build_generated(ctx, 1ll << scale, n_edges, flags, weight_seed, true, scale, seed,
                         a, b, c, out);
}

extern "C" mgx_status mgx_graph_uniform(mgx_context *ctx, int64_t n_vertices, int64_t n_edges,
                                        uint64_t seed, uint32_t flags, uint64_t weight_seed,
                                        mgx_graph **out) {
  return build_generated(ctx, n_vertices, n_edges, flags, weight_seed, false, 0, seed, 0, 0,
                         0, out);
}

extern "C" mgx_status mgx_graph_rmat_sharded(mgx_context *ctx, int scale, int64_t n_edges,
                                             uint64_t seed, double a, double b, double c,
                                             int64_t row_begin, int64_t row_end,
                                             mgx_graph **out) {
  const int64_t V = 1ll << scale;
  MGX_TRY(check_sizes(V, n_edges));
  if (row_begin < 0 || row_end < row_begin) {
    mgx_set_error("bad shard range [%lld,%lld)", (long long)row_begin, (long long)row_end);
    return MGX_ERR_INVALID_ARGUMENT;
  }
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  int32_t *d_src = nullptr, *d_dst = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_src, (n_edges > 0 ? n_edges : 1) * sizeof(int32_t)));
  MGX_HIP_TRY(mgx_hip_malloc(&d_dst, (n_edges > 0 ? n_edges : 1) * sizeof(int32_t)));
  mgx_status s = mgx_gen_rmat_device(ctx, scale, n_edges, seed, a, b, c, d_src, d_dst);
  mgx_graph *g = nullptr;
  if (s == MGX_OK) {
    g = new mgx_graph();
    const int64_t clamped_end = row_end < V ? row_end : V;
    s = mgx_build_sharded_in_csr(ctx, d_src, d_dst, V, n_edges, row_begin, clamped_end, g);
    if (s == MGX_OK) {
      g->row_end = row_end;  // keep the padded shard size for allgather
    }
  }
  (void)hipFree(d_src);
  (void)hipFree(d_dst);
  if (s != MGX_OK) {
    if (g) (void)mgx_graph_destroy(ctx, g);
    return s;
  }
  *out = g;
  return MGX_OK;
}

extern "C" mgx_status mgx_graph_destroy(mgx_context *ctx, mgx_graph *g) {
  if (!g) return MGX_OK;
  (void)ctx;
  if (g->in_row_ptr) (void)hipFree(g->in_row_ptr);
  if (g->in_col) (void)hipFree(g->in_col);
  if (g->out_degree) (void)hipFree(g->out_degree);
  if (g->inv_outdeg) (void)hipFree(g->inv_outdeg);
  if (g->out_row_ptr) (void)hipFree(g->out_row_ptr);
  if (g->out_col) (void)hipFree(g->out_col);
  if (g->in_w) (void)hipFree(g->in_w);
  if (g->bins_out.rows) (void)hipFree(g->bins_out.rows);
  if (g->sym_row_ptr) (void)hipFree(g->sym_row_ptr);
  if (g->sym_col) (void)hipFree(g->sym_col);
  if (g->sym_w) (void)hipFree(g->sym_w);
  if (g->bins_in.rows) (void)hipFree(g->bins_in.rows);
  if (g->bins_sym.rows) (void)hipFree(g->bins_sym.rows);
  if (g->stripe_ptr) (void)hipFree(g->stripe_ptr);
  if (g->order) (void)hipFree(g->order);
  for (int i = 0; i < 16; ++i) {
    if (g->stripe_bins[i].rows) (void)hipFree(g->stripe_bins[i].rows);
  }
  delete g;
  return MGX_OK;
}

extern "C" int64_t mgx_graph_num_vertices(const mgx_graph *g) { return g->n_vertices; }
extern "C" int64_t mgx_graph_num_edges(const mgx_graph *g) { return g->n_edges; }
extern "C" int64_t mgx_graph_local_edges(const mgx_graph *g) { return g->in_edges; }
extern "C" double mgx_graph_build_ms(const mgx_graph *g) { return g->build_ms; }

extern "C" mgx_status mgx_wcc(mgx_context *ctx, mgx_graph *g, int64_t *out_component,
                              int64_t *n_components) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  return mgx_wcc_impl(ctx, g, out_component, n_components);
}

extern "C" mgx_status mgx_katz(mgx_context *ctx, mgx_graph *g, double alpha, double epsilon,
                               double *out_centrality, int64_t *iterations) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  return mgx_katz_impl(ctx, g, alpha, epsilon, out_centrality, iterations);
}

extern "C" mgx_status mgx_louvain(mgx_context *ctx, mgx_graph *g, double threshold,
                                  int64_t *out_community, int64_t *n_communities) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  return mgx_louvain_impl(ctx, g, threshold, out_community, n_communities);
}

extern "C" mgx_status mgx_betweenness(mgx_context *ctx, mgx_graph *g, int directed,
                                      int normalize, double *out_bc) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  return mgx_betweenness_impl(ctx, g, directed, normalize, out_bc);
}
