// Internal structures of libmgx_analytics (gfx950-only HIP).
#ifndef MGX_INTERNAL_H
#define MGX_INTERNAL_H

#include <hip/hip_runtime.h>

#include <cstdint>
#include <cstdio>
#include <map>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include "../../include/mgx_analytics.h"

// Set per-thread error detail and return a status.
void mgx_set_error(const char *fmt, ...);

#define MGX_HIP_TRY(expr)                                                     \
  do {                                                                        \
    hipError_t _e = (expr);                                                   \
    if (_e != hipSuccess) {                                                   \
      mgx_set_error("%s:%d: %s failed: %s", __FILE__, __LINE__, #expr,        \
                    hipGetErrorString(_e));                                   \
      (void)hipGetLastError(); /* consume the sticky per-thread error so a  \
                                  later rocprim dispatch doesn't inherit it */ \
      return MGX_ERR_HIP;                                                     \
    }                                                                         \
  } while (0)

#define MGX_TRY(expr)                                                         \
  do {                                                                        \
    mgx_status _s = (expr);                                                   \
    if (_s != MGX_OK) return _s;                                              \
  } while (0)

struct mgx_comm_state;  // comm.cpp (RCCL)

// The box's ROCm 7.0.x runtime mis-reserves plain hipMalloc requests that
// are >4 GiB and NOT a multiple of 2^32 bytes: the returned range overlaps
// live allocations and writing past the low 32 bits of the size faults
// ("write to a read-only page"). Every multi-GB allocation that ever
// worked here was an exact 2^32 multiple (power-of-two element counts);
// the first non-aligned huge malloc (Louvain coarsen's total-sized w at
// RMAT-25) hit the bug. Round such sizes up to the next 2^32 multiple.
inline size_t mgx_safe_size(size_t bytes) {
  if (bytes >= (4ull << 30) && (bytes & 0xFFFFFFFFull) != 0)
    bytes = (bytes + 0xFFFFFFFFull) & ~0xFFFFFFFFull;
  return bytes;
}

template <typename T>
inline hipError_t mgx_hip_malloc(T **p, size_t bytes) {
  return hipMalloc((void **)p, mgx_safe_size(bytes));
}

struct mgx_context {
  int device = -1;
  hipStream_t stream = nullptr;
  // Caching allocator for large transient buffers (sort keys, COO staging):
  // a size-bucketed free list over plain hipMalloc, so repeated multi-GB
  // allocation page-table setup (measured up to ~1 s per RMAT-26 build) is
  // paid once per process. Reuse is safe without syncs because every
  // consumer enqueues on ctx->stream: the previous owner's work precedes
  // the next owner's in stream order.
  //
  // Deliberately NOT hipMallocAsync: on the MI355X box's ROCm 7.0.x runtime
  // a plain hipMalloc issued while the stream-ordered pool holds
  // pending-free-then-reallocated blocks reclaims (and zeroes) the live
  // blocks' pages — measured in the r02 Louvain storm (counts[] nonzero,
  // then hipMalloc(row_ptr), then counts[] all-zero at disjoint VA; see
  // profiles/r02_summary.md). One allocator family avoids the bug class.
  std::mutex cache_mu;
  std::unordered_map<void *, size_t> cache_live;  // ptr -> rounded bytes
  std::multimap<size_t, void *> cache_free;       // rounded bytes -> ptr
  size_t cache_free_bytes = 0;
  mgx_status alloc_async(void **p, size_t bytes);
  mgx_status free_async(void *p);
  void cache_trim();  // hipFree every cached free block
  // Release the cache if a coming plain hipMalloc of `want` bytes would
  // leave the device under the safety margin (see alloc_async).
  void ensure_margin(size_t want);
  // Grow-only device workspace for rocPRIM temp storage etc.
  void *workspace = nullptr;
  size_t workspace_bytes = 0;
  // Louvain coarsen col/w ping-pong arenas. The ROCm 7.0.x runtime
  // misplaces >4 GiB hipMallocs once the VA space is fragmented
  // (profiles/r02_summary.md), and the coarsen's total-sized col/w are
  // the only such allocations issued mid-run — so they come from these
  // grow-only arenas, sized and allocated at mgx_louvain entry before
  // any per-level churn. [2] because level k's graph must stay live
  // while level k+1 is being written.
  void *louv_col[2] = {nullptr, nullptr};
  void *louv_w[2] = {nullptr, nullptr};
  size_t louv_col_bytes[2] = {0, 0};
  size_t louv_w_bytes[2] = {0, 0};
  int louv_flip = 0;
  mgx_comm_state *comm = nullptr;

  mgx_status reserve(size_t bytes, void **out);
};

// Degree bins for the fused sweep kernels. Rows are classified by degree;
// within a bin rows are in ascending id order (stable 2-bit radix sort) so
// adjacent lanes read adjacent CSR ranges.
//   bin0: deg < 8     -> 4 lanes/row
//   bin1: 8..63       -> 16 lanes/row
//   bin2: 64..1023    -> 64 lanes (one wave)/row
//   bin3: >= 1024     -> one 256-thread workgroup/row
struct mgx_bins {
  int32_t *rows = nullptr;  // [n_rows] device: bin0 rows, then bin1, ...
  int64_t count[4] = {0, 0, 0, 0};
  // Launch geometry (computed host-side at build).
  int64_t grid[4] = {0, 0, 0, 0};
  int64_t total_grid() const { return grid[0] + grid[1] + grid[2] + grid[3]; }
};

struct mgx_graph {
  int64_t n_vertices = 0;
  int64_t n_edges = 0;
  uint32_t flags = 0;

  // in-CSR: row = destination, cols = sources (PageRank/Katz pull).
  // For sharded graphs, rows cover [row_begin, row_end) and in_row_ptr has
  // (row_end - row_begin + 1) entries; in_col holds only owned edges.
  uint32_t *in_row_ptr = nullptr;
  int32_t *in_col = nullptr;
  int64_t in_edges = 0;  // == n_edges unless sharded
  int64_t row_begin = 0, row_end = 0;  // == [0, n_vertices) unless sharded

  // out-degree data (always global, all vertices). When `order` is set,
  // the in-CSR lives in a HOT-FIRST PERMUTED vertex space: vertices are
  // renumbered by descending out-degree so the most-gathered contrib
  // entries pack into the lowest addresses (L2/L3-resident under power-law
  // skew). order[new] = original id; out_degree/inv_outdeg are in the
  // permuted space. Outputs are scattered back to original ids at download.
  // Sharded graphs keep the identity layout (contiguous dst ranges).
  uint32_t *out_degree = nullptr;
  float *inv_outdeg = nullptr;
  int32_t *order = nullptr;  // [V] new -> original, or nullptr (identity)

  // out-CSR (row = source, cols = destinations; identity layout) for
  // directed Brandes BFS.
  uint32_t *out_row_ptr = nullptr;
  int32_t *out_col = nullptr;

  // in-CSR edge weights, built only with MGX_BUILD_NO_PERM|WEIGHTED
  // (directed LabelRankT).
  float *in_w = nullptr;

  // symmetric CSR (WCC/Louvain): each input edge twice.
  uint32_t *sym_row_ptr = nullptr;
  int32_t *sym_col = nullptr;
  float *sym_w = nullptr;  // only with MGX_BUILD_WEIGHTED

  mgx_bins bins_in;   // over in-CSR rows
  mgx_bins bins_sym;  // over sym-CSR rows
  mgx_bins bins_out;  // over out-CSR rows

  // Source-striped in-CSR view (PageRank at large V): cols are sorted
  // within each row, so stripe s of row v is the contiguous sub-range
  // [stripe_ptr[s*rows + v], stripe_ptr[(s+1)*rows + v]) whose sources lie
  // in [s*stripe_width, (s+1)*stripe_width) — the gathered contrib stripe
  // then fits in the Infinity Cache. n_stripes == 1 => unstriped.
  int n_stripes = 1;
  int64_t stripe_width = 0;
  uint32_t *stripe_ptr = nullptr;          // [(n_stripes+1) * rows]
  mgx_bins stripe_bins[16];                // per stripe (max 16)

  double build_ms = 0.0;
};

// graph_build.hip entry points (device COO is int32).
mgx_status mgx_build_from_device_coo(mgx_context *ctx, const int32_t *d_src,
                                     const int32_t *d_dst, const float *d_w,
                                     int64_t n_vertices, int64_t n_edges, uint32_t flags,
                                     mgx_graph *g);
mgx_status mgx_gen_rmat_device(mgx_context *ctx, int scale, int64_t n_edges, uint64_t seed,
                               double a, double b, double c, int32_t *d_src, int32_t *d_dst);
mgx_status mgx_gen_uniform_device(mgx_context *ctx, int64_t n_vertices, int64_t n_edges,
                                  uint64_t seed, int32_t *d_src, int32_t *d_dst);
mgx_status mgx_gen_weights_device(mgx_context *ctx, int64_t n_edges, uint64_t seed, float *d_w);
// Build the degree-bin work lists for a CSR with `rows` rows.
mgx_status mgx_build_bins(mgx_context *ctx, const uint32_t *row_ptr, int64_t rows,
                          mgx_bins *bins);
// Degree-bin work lists from explicit per-row [lo, hi) ranges; rows with
// zero degree are dropped unless include_zero.
mgx_status mgx_build_bins_range(mgx_context *ctx, const uint32_t *lo, const uint32_t *hi,
                                int64_t rows, bool include_zero, mgx_bins *bins);
// Source-stripe the (sorted) in-CSR of g; n_stripes chosen from V (env
// MGX_PR_STRIPES overrides; 1 = disabled).
mgx_status mgx_build_stripes(mgx_context *ctx, mgx_graph *g);

// Sharded in-CSR build: keeps only edges with dst in [row_begin,row_end);
// out_degree stays global.
mgx_status mgx_build_sharded_in_csr(mgx_context *ctx, const int32_t *d_src,
                                    const int32_t *d_dst, int64_t n_vertices,
                                    int64_t n_edges, int64_t row_begin, int64_t row_end,
                                    mgx_graph *g);

// pagerank.hip
struct mgx_pagerank_run {
  mgx_context *ctx = nullptr;
  mgx_graph *g = nullptr;
  double damping = 0.85;
  float *rank[2] = {nullptr, nullptr};     // ping-pong f32 [V]
  float *contrib[2] = {nullptr, nullptr};  // ping-pong f32 [V]
  int cur = 0;
  uint32_t *d_delta = nullptr;  // Linf as ordered-uint f32
  double *d_scratch = nullptr;  // sum + f64 output [V+1]
  double *d_partial = nullptr;  // striped sweep: fp64 row partials [rows]
  int64_t iterations = 0;
  // HIP-event timing of the sweep kernel (the dominant kernel).
  std::vector<hipEvent_t> ev_start, ev_stop;
  int64_t ev_used = 0;
  double sweep_ms_acc = 0.0;
  int64_t launches_acc = 0;
  // distributed
  bool dist = false;
  int64_t row_begin = 0, row_end = 0;

  mgx_status flush_timing();  // sync + fold events into sweep_ms_acc
};

mgx_status mgx_pagerank_queue_iterations(mgx_pagerank_run *run, int64_t n, bool track_delta);
mgx_status mgx_pagerank_read_delta(mgx_pagerank_run *run, float *out);  // syncs
mgx_status mgx_pagerank_normalize_download(mgx_pagerank_run *run, double *out_rank);

// wcc.hip
mgx_status mgx_wcc_impl(mgx_context *ctx, mgx_graph *g, int64_t *out_component,
                        int64_t *n_components);
// katz.hip
mgx_status mgx_katz_impl(mgx_context *ctx, mgx_graph *g, double alpha, double epsilon,
                         double *out_centrality, int64_t *iterations);
// louvain.hip
mgx_status mgx_louvain_impl(mgx_context *ctx, mgx_graph *g, double threshold,
                            int64_t *out_community, int64_t *n_communities);
// betweenness.hip
mgx_status mgx_betweenness_impl(mgx_context *ctx, mgx_graph *g, int directed, int normalize,
                                double *out_bc);

// comm.cpp (RCCL)
mgx_status mgx_comm_allgather_f32(mgx_context *ctx, const float *send, float *recv,
                                  size_t per_rank_count);
mgx_status mgx_comm_allreduce_max_f32(mgx_context *ctx, const float *send, float *recv);
int mgx_comm_world(mgx_context *ctx);  // 0 if no comm

#endif  // MGX_INTERNAL_H
