// Leiden community detection on gfx950 — replaces leiden_alg
// (/root/reference/src/mage/cpp/leiden_community_detection_module/algorithm/
// leiden.cpp):
//   graph intake (:489-516): undirected SIMPLE graph — parallel edges are
//     dropped keeping the FIRST-scanned weight (edge_exists skip :504-507);
//     self edges contribute two adjacency entries; node_weight = sum of
//     incident kept weights; gamma /= sum_of_weights (:527).
//   MoveNodesFast (:52-135): CPM local moves — delta(C) = w(v->C) -
//     |C|*gamma, move when delta > best + resolution_parameter.
//   RefinePartition/MergeNodesSubset (:185-336): within each community,
//     singleton well-connected nodes merge into well-connected refined
//     communities; the target is drawn from exp(delta/theta) cumulative
//     weights via lower_bound — including the reference's always-merge
//     behavior when every delta <= resolution (the cumsum is all zeros and
//     lower_bound lands on the first neighbour community; the
//     `total_cum_sum < MAX_DOUBLE` guard at :268 is always true).
//   AggregateGraph (:357-446): refined communities become nodes; ONE
//     representative edge per community pair (first encountered) carries
//     its weight; partitions group refined communities by the original
//     partition; dendrogram levels accumulate (leiden_utils.cpp:19-30).
//   Main loop (:529-563): stop on all-singleton, everything-merged
//     (refined empties == size-1), no-refinement special branch, or
//     max_iterations.
//
// The reference is RANDOMIZED (std::random_device-seeded shuffle in
// MoveNodesFast :65 and minstd draw in MergeNodesSubset :268) and its
// trajectories depend on queue order, so parity is the DESIGN.md
// statistical bar: exact hierarchy equality on trajectory-stable golden
// graphs, and partition-quality (CPM/modularity, community counts) within
// the reference run distribution elsewhere. Documented scheduling
// divergences: MoveNodesFast runs as red-black (node-id parity) Jacobi
// sweeps to a fixed point instead of the sequential shuffled queue;
// refinement processes subsets in parallel with members in ascending node
// order; the aggregated pair representative is the min-scan-order edge.
// RNG is counter-based splitmix64 keyed (seed, level, subset).
//
// Device layout per level: simple undirected CSR (u32 row_ptr, i32 col,
// f64 w), fp64 node weights, i32 community ids + u32 community sizes.
// Sweeps reuse the louvain.hip shape: LDS open-addressing tables for rows
// under 256 neighbours, a global pool region per hub row.

#include <cstring>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include <rocprim/rocprim.hpp>

#include "../../include/mgx_graphgen.h"
#include "mgx_internal.h"

namespace {

// MGX_LEIDEN_TRACE=1: per-level phase markers on stderr (diagnostics only).
inline bool leiden_trace() {
  static const bool v = [] {
    const char *e = getenv("MGX_LEIDEN_TRACE");
    return e && atoi(e) != 0;
  }();
  return v;
}
#define MGX_LEIDEN_LOG(...)                \
  do {                                     \
    if (leiden_trace()) {                  \
      fprintf(stderr, "[leiden] " __VA_ARGS__); \
      fprintf(stderr, "\n");               \
      fflush(stderr);                      \
    }                                      \
  } while (0)

constexpr int kBlock = 256;
constexpr uint32_t kSmallRowDeg = 256;
constexpr int kLdsCap = 512;

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

struct Level {
  int64_t nv = 0;
  uint32_t *row_ptr = nullptr;
  int32_t *col = nullptr;
  double *w = nullptr;
  double *node_w = nullptr;
  int64_t ne2 = 0;
};

void free_level(Level *L) {
  if (L->row_ptr) (void)hipFree(L->row_ptr);
  if (L->col) (void)hipFree(L->col);
  if (L->w) (void)hipFree(L->w);
  if (L->node_w) (void)hipFree(L->node_w);
  *L = Level{};
}

// ---- generic kernels ----------------------------------------------------

__global__ void k_iota32(int64_t n, int32_t *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = (int32_t)i;
}

__global__ void k_fill32(int64_t n, int32_t v, int32_t *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = v;
}

__global__ void k_fillu32(int64_t n, uint32_t v, uint32_t *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = v;
}

// ---- MoveNodesFast as red-black CPM sweeps ------------------------------

struct MoveArgs {
  const uint32_t *row_ptr;
  const int32_t *col;
  const double *w;
  const int32_t *comm;    // snapshot
  const uint32_t *csize;  // community sizes (counts)
  int32_t *target;
  double gamma;
  double resolution;
  int parity;  // process nodes with (id & 1) == parity
  // hub pool
  const int32_t *small_rows;
  int64_t n_small;
  const int32_t *big_rows;
  int64_t n_big;
  int32_t *pool_keys;
  double *pool_vals;
  const uint64_t *pool_off;
};

__device__ inline void leiden_move_finish(const MoveArgs &A, int32_t v, int32_t best) {
  A.target[v] = best;
}

// wave per small row, LDS table of (community, weight)
__global__ void __launch_bounds__(kBlock) k_move_small(MoveArgs A) {
  __shared__ int32_t keys[kBlock / 64][kLdsCap];
  __shared__ double vals[kBlock / 64][kLdsCap];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t waves_per_grid = (int64_t)gridDim.x * (kBlock / 64);
  for (int64_t wi = (int64_t)blockIdx.x * (kBlock / 64) + wave; wi < A.n_small;
       wi += waves_per_grid) {
    const int32_t v = A.small_rows[wi];
    if ((v & 1) != A.parity) continue;
    const uint32_t s = A.row_ptr[v], e = A.row_ptr[v + 1];
    const int32_t cv = A.comm[v];
    for (int t = lane; t < kLdsCap; t += 64) {
      keys[wave][t] = -1;
      vals[wave][t] = 0.0;
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    for (uint32_t j = s + lane; j < e; j += 64) {
      const int32_t c = A.comm[A.col[j]];
      uint32_t h = ((uint32_t)c * 2654435761u) & (kLdsCap - 1);
      while (true) {
        int32_t prev = atomicCAS(&keys[wave][h], -1, c);
        if (prev == -1 || prev == c) break;
        h = (h + 1) & (kLdsCap - 1);
      }
      atomicAdd(&vals[wave][h], A.w[j]);
    }
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
    // current delta: w(v->C_v) - (|C_v|-1)*gamma (:88-90)
    double wcv = 0.0;
    {
      uint32_t h = ((uint32_t)cv * 2654435761u) & (kLdsCap - 1);
      for (int probe = 0; probe < kLdsCap; ++probe) {
        const int32_t k = keys[wave][h];
        if (k == cv) {
          wcv = vals[wave][h];
          break;
        }
        if (k == -1) break;
        h = (h + 1) & (kLdsCap - 1);
      }
    }
    double best_delta = wcv - ((double)A.csize[cv] - 1.0) * A.gamma;
    int32_t best = cv;
    for (int t = lane; t < kLdsCap; t += 64) {
      const int32_t c = keys[wave][t];
      if (c >= 0 && c != cv) {
        const double delta = vals[wave][t] - (double)A.csize[c] * A.gamma;
        if (delta > best_delta + A.resolution ||
            (delta == best_delta + A.resolution && false)) {
          best_delta = delta;
          best = c;
        }
      }
    }
    // wave argmax (deterministic tie-break: larger delta, then smaller id)
    for (int o = 32; o; o >>= 1) {
      const double od = __shfl_down(best_delta, o, 64);
      const int32_t ob = __shfl_down(best, o, 64);
      if (od > best_delta || (od == best_delta && ob < best)) {
        best_delta = od;
        best = ob;
      }
    }
    if (lane == 0) leiden_move_finish(A, v, best);
    asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
  }
}

__global__ void __launch_bounds__(kBlock) k_move_big(MoveArgs A) {
  __shared__ double s_d[kBlock / 64];
  __shared__ int32_t s_b[kBlock / 64];
  for (int64_t bi = blockIdx.x; bi < A.n_big; bi += gridDim.x) {
    const int32_t v = A.big_rows[bi];
    if ((v & 1) != A.parity) continue;
    const uint32_t s = A.row_ptr[v], e = A.row_ptr[v + 1];
    const int32_t cv = A.comm[v];
    const uint64_t t0 = A.pool_off[bi], t1 = A.pool_off[bi + 1];
    const uint32_t cap = (uint32_t)(t1 - t0);
    int32_t *keys = A.pool_keys + t0;
    double *vals = A.pool_vals + t0;
    for (uint32_t t = threadIdx.x; t < cap; t += kBlock) {
      keys[t] = -1;
      vals[t] = 0.0;
    }
    __syncthreads();
    for (uint32_t j = s + threadIdx.x; j < e; j += kBlock) {
      const int32_t c = A.comm[A.col[j]];
      uint32_t h = ((uint32_t)c * 2654435761u) & (cap - 1);
      while (true) {
        int32_t prev = atomicCAS(&keys[h], -1, c);
        if (prev == -1 || prev == c) break;
        h = (h + 1) & (cap - 1);
      }
      atomicAdd(&vals[h], A.w[j]);
    }
    __syncthreads();
    double wcv = 0.0;
    {
      uint32_t h = ((uint32_t)cv * 2654435761u) & (cap - 1);
      for (uint32_t probe = 0; probe < cap; ++probe) {
        const int32_t k = keys[h];
        if (k == cv) {
          wcv = vals[h];
          break;
        }
        if (k == -1) break;
        h = (h + 1) & (cap - 1);
      }
    }
    double best_delta = wcv - ((double)A.csize[cv] - 1.0) * A.gamma;
    int32_t best = cv;
    for (uint32_t t = threadIdx.x; t < cap; t += kBlock) {
      const int32_t c = keys[t];
      if (c >= 0 && c != cv) {
        const double delta = vals[t] - (double)A.csize[c] * A.gamma;
        if (delta > best_delta + A.resolution) {
          best_delta = delta;
          best = c;
        }
      }
    }
    for (int o = 32; o; o >>= 1) {
      const double od = __shfl_down(best_delta, o, 64);
      const int32_t ob = __shfl_down(best, o, 64);
      if (od > best_delta || (od == best_delta && ob < best)) {
        best_delta = od;
        best = ob;
      }
    }
    if ((threadIdx.x & 63) == 0) {
      s_d[threadIdx.x >> 6] = best_delta;
      s_b[threadIdx.x >> 6] = best;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      for (int i = 1; i < kBlock / 64; ++i) {
        if (s_d[i] > best_delta || (s_d[i] == best_delta && s_b[i] < best)) {
          best_delta = s_d[i];
          best = s_b[i];
        }
      }
      leiden_move_finish(A, v, best);
    }
    __syncthreads();
  }
}

// apply targets for parity nodes; update sizes; count moves
__global__ void k_move_apply(int64_t nv, int parity, const int32_t *target, int32_t *comm,
                             uint32_t *csize, unsigned long long *n_moved) {
  for (int64_t v = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; v < nv;
       v += (int64_t)gridDim.x * blockDim.x) {
    if ((v & 1) != parity) continue;
    const int32_t t = target[v];
    const int32_t c = comm[v];
    if (t != c) {
      atomicSub(&csize[c], 1u);
      atomicAdd(&csize[t], 1u);
      comm[v] = t;
      atomicAdd(n_moved, 1ull);
    }
  }
}

__global__ void k_count_nonsingleton(int64_t nv, const uint32_t *csize,
                                     unsigned long long *out_nonsingleton,
                                     unsigned long long *out_nonempty) {
  unsigned long long ns = 0, ne = 0;
  for (int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; c < nv;
       c += (int64_t)gridDim.x * blockDim.x) {
    if (csize[c] > 1) ++ns;
    if (csize[c] > 0) ++ne;
  }
  if (ns) atomicAdd(out_nonsingleton, ns);
  if (ne) atomicAdd(out_nonempty, ne);
}

// ---- refinement ---------------------------------------------------------

// One thread per subset (original community); members are the sorted-by-
// community node list slice [mem_off[c], mem_off[c+1]). Local refined
// community = member index within the subset. Scratch arrays are slices of
// global pools at mem_off[c].
struct RefineArgs {
  const uint32_t *row_ptr;
  const int32_t *col;
  const double *w;
  const double *node_w;
  const int32_t *comm;      // original partition
  const int32_t *members;   // nodes sorted by comm
  const uint32_t *mem_off;  // [n_comm+1]
  int64_t n_comm;
  const int32_t *loc_idx;  // node -> index within its subset
  int32_t *rcomm_loc;      // node -> local refined community (member idx)
  // scratch pools sliced per subset (size = subset size each)
  double *ext_w;
  double *edge_w;
  int32_t *nb_comms;
  int32_t *nb_slot;  // rc -> 1+index into nb_comms, 0 = absent (O(1) lookup)
  double *prob;
  uint32_t *rsize;  // refined community local sizes
  double gamma;
  double theta;
  double resolution;
  uint64_t seed;  // mixed, per level
  unsigned long long *n_merged;
};

__global__ void k_refine(RefineArgs A) {
  for (int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; c < A.n_comm;
       c += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t m0 = A.mem_off[c], m1 = A.mem_off[c + 1];
    const uint32_t k = m1 - m0;
    if (k <= 1) continue;  // only communities larger than one (:323-325)
    double *ext_w = A.ext_w + m0;
    double *edge_w = A.edge_w + m0;
    int32_t *nbc = A.nb_comms + m0;
    int32_t *nslot = A.nb_slot + m0;
    double *prob = A.prob + m0;
    uint32_t *rsize = A.rsize + m0;
    for (uint32_t i = 0; i < k; ++i) {
      ext_w[i] = 0.0;
      edge_w[i] = 0.0;
      nslot[i] = 0;
      rsize[i] = 1;
      A.rcomm_loc[A.members[m0 + i]] = (int32_t)i;
    }
    // initial external weights + subset weight (:196-203)
    double subset_weight = 0.0;
    for (uint32_t i = 0; i < k; ++i) {
      const int32_t v = A.members[m0 + i];
      for (uint32_t j = A.row_ptr[v]; j < A.row_ptr[v + 1]; ++j) {
        if (A.comm[A.col[j]] == (int32_t)c) ext_w[i] += A.w[j];
      }
      subset_weight += A.node_w[v];
    }
    uint64_t draw = 0;
    for (uint32_t i = 0; i < k; ++i) {
      const int32_t v = A.members[m0 + i];
      const int32_t cur = A.rcomm_loc[v];
      const double node_weight = A.node_w[v];
      // singleton + well-connected gate (:215-221)
      if (rsize[cur] != 1) continue;
      const double right = A.gamma * node_weight * (subset_weight - node_weight);
      if (!(ext_w[cur] >= right)) continue;
      // neighbour refined communities + weights (:228-238)
      int32_t nn = 0;
      for (uint32_t j = A.row_ptr[v]; j < A.row_ptr[v + 1]; ++j) {
        const int32_t u = A.col[j];
        if (A.comm[u] != (int32_t)c) continue;
        const int32_t rc = A.rcomm_loc[u];
        // First-encounter append, O(1) via the slot table (semantically
        // identical to a linear membership scan of nbc[0..nn), which is
        // O(deg^2) for hub members and hung RMAT-scale runs).
        if (edge_w[rc] == 0.0 && nslot[rc] == 0) {
          nbc[nn] = rc;
          nslot[rc] = nn + 1;
          ++nn;
        }
        edge_w[rc] += A.w[j];
      }
      double total = 0.0;
      double max_delta = 0.0;
      int32_t best = cur;
      for (int32_t q = 0; q < nn; ++q) {
        const int32_t rc = nbc[q];
        double p = total;
        if (rsize[rc] != 0) {
          // well-connected community gate (:245-251); community weight is
          // the COUNT (GetCommunityWeight = size), subset term uses the
          // subset's node count (:248-250)
          const double rightc =
              A.gamma * (double)rsize[rc] * ((double)k - (double)rsize[rc]);
          if (ext_w[rc] >= rightc) {
            const double delta = edge_w[rc] - (double)rsize[rc] * A.gamma;
            if (delta > A.resolution) total += exp(delta / A.theta);
            if (delta > max_delta + A.resolution) {
              max_delta = delta;
              best = rc;
            }
          }
        }
        p = total;
        prob[q] = p;  // cumulative snapshot per neighbour slot (:264)
        edge_w[rc] = 0.0;
        nslot[rc] = 0;
      }
      if (nn > 0) {
        // the reference's always-true branch (:268-276): draw in [0,total)
        // and lower_bound over the cumulative snapshots — with an all-zero
        // cumsum this lands on the first neighbour community
        const uint64_t h = mgx_hash64(A.seed, ((uint64_t)c << 20) ^ draw);
        ++draw;
        const double u01 = (double)(h >> 11) * (1.0 / 9007199254740992.0);
        const double r = u01 * total;
        int32_t pick = 0;
        while (pick < nn && prob[pick] < r) ++pick;
        if (pick >= nn) pick = nn - 1;
        best = nbc[pick];
      }
      if (best != cur) {
        A.rcomm_loc[v] = best;
        rsize[best] += 1;
        rsize[cur] = 0;
        atomicAdd(A.n_merged, 1ull);
        // External weights stay STALE after a merge: the reference's
        // post-merge update (:317-327) guards on
        // `refined_community == subset`, comparing ids from two different
        // id spaces, so it is a de-facto no-op in almost every execution.
        // Replicating our own ids through that confused condition is
        // meaningless; we replicate the dominant (stale) behavior and the
        // statistical bar covers the residual divergence.
      }
    }
  }
}

}  // namespace

// host-side per-level state for aggregation/dendrogram is orchestrated in
// mgx_leiden below with host copies (the per-level arrays are nv-sized).

extern "C" mgx_status mgx_leiden(mgx_context *ctx, mgx_graph *g, double gamma,
                                 double theta, double resolution, int64_t max_iterations,
                                 uint64_t seed, int64_t cap,
                                 int64_t *out_hier /* [V*cap], -1 pad */,
                                 int64_t *out_levels /* [V] */) {
  if (g && !(g->flags & MGX_BUILD_SYM_CSR)) {
    mgx_set_error("leiden needs SYM_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  const int64_t V = g ? g->n_vertices : 0;
  for (int64_t v = 0; v < V; ++v) {
    out_levels[v] = 0;
    for (int64_t k = 0; k < cap; ++k) out_hier[v * cap + k] = -1;
  }
  if (V == 0 || g->n_edges == 0) return MGX_OK;  // Leiden returns {} (:485-487)

  // ---- level-0 simple graph from the (stable-sorted) sym CSR ------------
  // Keep the FIRST entry of each (row,col) run — the first-scanned edge —
  // and two entries for self runs (AddEdge both directions :508-512).
  const int64_t ne2_in = 2 * g->n_edges;
  std::vector<uint32_t> rp_h(V + 1);
  MGX_HIP_TRY(hipMemcpyAsync(rp_h.data(), g->sym_row_ptr, (V + 1) * 4,
                             hipMemcpyDeviceToHost, ctx->stream));
  std::vector<int32_t> col_h(ne2_in);
  MGX_HIP_TRY(hipMemcpyAsync(col_h.data(), g->sym_col, ne2_in * 4, hipMemcpyDeviceToHost,
                             ctx->stream));
  std::vector<float> w_h;
  if (g->sym_w) {
    w_h.resize(ne2_in);
    MGX_HIP_TRY(hipMemcpyAsync(w_h.data(), g->sym_w, ne2_in * 4, hipMemcpyDeviceToHost,
                               ctx->stream));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  // host-side dedup (one pass over E; the level-0 build is not a hot path —
  // the sweeps are)
  std::vector<uint32_t> rp2(V + 1, 0);
  std::vector<int32_t> col2;
  std::vector<double> w2;
  col2.reserve(ne2_in);
  w2.reserve(ne2_in);
  std::vector<double> nodew_h(V, 0.0);
  double sum_of_weights = 0.0;
  for (int64_t v = 0; v < V; ++v) {
    int32_t run = -1;
    int run_cnt = 0;
    for (uint32_t j = rp_h[v]; j < rp_h[v + 1]; ++j) {
      const int32_t c = col_h[j];
      if (c != run) {
        run = c;
        run_cnt = 0;
      }
      ++run_cnt;
      const int keep = (c == v) ? 2 : 1;
      if (run_cnt <= keep) {
        const double wj = g->sym_w ? (double)w_h[j] : 1.0;
        col2.push_back(c);
        w2.push_back(wj);
        nodew_h[v] += wj;
        if (c >= v) sum_of_weights += (c == v) ? wj * 0.5 * (run_cnt == 1 ? 1.0 : 1.0)
                                               : wj;
      }
    }
    rp2[v + 1] = (uint32_t)col2.size();
  }
  // self-edge sum correction: each kept self edge appears twice with w each;
  // the reference counts it once in sum_of_weights. The loop above added
  // 0.5*w per kept self entry (two entries -> w total). Non-self edges were
  // added once (c >= v).
  const double gamma_n = gamma / sum_of_weights;

  Level L;
  L.nv = V;
  L.ne2 = (int64_t)col2.size();
  MGX_HIP_TRY(mgx_hip_malloc(&L.row_ptr, (V + 1) * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&L.col, (L.ne2 > 0 ? L.ne2 : 1) * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&L.w, (L.ne2 > 0 ? L.ne2 : 1) * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&L.node_w, V * 8));
  MGX_HIP_TRY(hipMemcpyAsync(L.row_ptr, rp2.data(), (V + 1) * 4, hipMemcpyHostToDevice,
                             ctx->stream));
  if (L.ne2 > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(L.col, col2.data(), L.ne2 * 4, hipMemcpyHostToDevice,
                               ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(L.w, w2.data(), L.ne2 * 8, hipMemcpyHostToDevice,
                               ctx->stream));
  }
  MGX_HIP_TRY(hipMemcpyAsync(L.node_w, nodew_h.data(), V * 8, hipMemcpyHostToDevice,
                             ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));

  // dendrogram: per level, mapping level-entity -> next-level community id
  std::vector<std::vector<int64_t>> dendro;

  const uint64_t ms = mgx_seed_mix(seed);
  int64_t level = 0;
  bool done = false;
  int64_t iters = 0;
  mgx_status st = MGX_OK;

  // partitions at level start: groups of current-level nodes. comm[] holds
  // it; at level 0 it is singleton.
  int32_t *d_comm = nullptr;
  uint32_t *d_csize = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_comm, V * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&d_csize, V * 4));
  hipLaunchKernelGGL(k_iota32, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, d_comm);
  hipLaunchKernelGGL(k_fillu32, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, 1u, d_csize);

  while (!done && st == MGX_OK) {
    ++iters;
    const int64_t nv = L.nv;
    MGX_LEIDEN_LOG("level %lld nv=%lld ne2=%lld", (long long)level, (long long)nv,
                   (long long)L.ne2);
    // ---- MoveNodesFast (red-black Jacobi to fixed point) ----------------
    // row classification for the sweep kernels
    std::vector<uint32_t> rp_l(nv + 1);
    MGX_HIP_TRY(hipMemcpyAsync(rp_l.data(), L.row_ptr, (nv + 1) * 4,
                               hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    std::vector<int32_t> srows, brows;
    std::vector<uint64_t> caps;
    uint64_t pool_total = 0;
    for (int64_t v = 0; v < nv; ++v) {
      const uint32_t d = rp_l[v + 1] - rp_l[v];
      if (d == 0) continue;
      if (d < kSmallRowDeg) {
        srows.push_back((int32_t)v);
      } else {
        brows.push_back((int32_t)v);
        uint64_t need = 2ull * d + 2;
        uint64_t cp = 1;
        while (cp < need) cp <<= 1;
        caps.push_back(pool_total);
        pool_total += cp;
      }
    }
    caps.push_back(pool_total);
    int32_t *d_srows = nullptr, *d_brows = nullptr, *d_target = nullptr;
    uint64_t *d_poff = nullptr;
    int32_t *d_pkeys = nullptr;
    double *d_pvals = nullptr;
    unsigned long long *d_cnt = nullptr;
    MGX_HIP_TRY(mgx_hip_malloc(&d_srows, (srows.empty() ? 1 : srows.size()) * 4));
    MGX_HIP_TRY(mgx_hip_malloc(&d_brows, (brows.empty() ? 1 : brows.size()) * 4));
    MGX_HIP_TRY(mgx_hip_malloc(&d_target, nv * 4));
    MGX_HIP_TRY(mgx_hip_malloc(&d_poff, caps.size() * 8));
    MGX_HIP_TRY(mgx_hip_malloc(&d_pkeys, (pool_total ? pool_total : 1) * 4));
    MGX_HIP_TRY(mgx_hip_malloc(&d_pvals, (pool_total ? pool_total : 1) * 8));
    MGX_HIP_TRY(mgx_hip_malloc(&d_cnt, 8));
    if (!srows.empty())
      MGX_HIP_TRY(hipMemcpyAsync(d_srows, srows.data(), srows.size() * 4,
                                 hipMemcpyHostToDevice, ctx->stream));
    if (!brows.empty())
      MGX_HIP_TRY(hipMemcpyAsync(d_brows, brows.data(), brows.size() * 4,
                                 hipMemcpyHostToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(d_poff, caps.data(), caps.size() * 8,
                               hipMemcpyHostToDevice, ctx->stream));

    unsigned long long empties_before = 0;
    {
      unsigned long long *d_t = nullptr;
      MGX_HIP_TRY(mgx_hip_malloc(&d_t, 16));
      MGX_HIP_TRY(hipMemsetAsync(d_t, 0, 16, ctx->stream));
      hipLaunchKernelGGL(k_count_nonsingleton, dim3((uint32_t)grid_for(nv)), dim3(kBlock),
                         0, ctx->stream, nv, d_csize, d_t, d_t + 1);
      unsigned long long h_t[2];
      MGX_HIP_TRY(hipMemcpyAsync(h_t, d_t, 16, hipMemcpyDeviceToHost, ctx->stream));
      MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
      empties_before = (unsigned long long)nv - h_t[1];
      (void)hipFree(d_t);
    }

    MoveArgs M;
    M.row_ptr = L.row_ptr;
    M.col = L.col;
    M.w = L.w;
    M.comm = d_comm;
    M.csize = d_csize;
    M.target = d_target;
    M.gamma = gamma_n;
    M.resolution = resolution;
    M.small_rows = d_srows;
    M.n_small = (int64_t)srows.size();
    M.big_rows = d_brows;
    M.n_big = (int64_t)brows.size();
    M.pool_keys = d_pkeys;
    M.pool_vals = d_pvals;
    M.pool_off = d_poff;
    int sweeps_done = 0;
    unsigned long long prev_moved = ~0ull;
    for (int sweep = 0; sweep < 200; ++sweep) {
      ++sweeps_done;
      unsigned long long moved_total = 0;
      for (int parity = 0; parity < 2; ++parity) {
        M.parity = parity;
        MGX_HIP_TRY(hipMemcpyAsync(d_target, d_comm, nv * 4, hipMemcpyDeviceToDevice,
                                   ctx->stream));
        if (!srows.empty())
          hipLaunchKernelGGL(k_move_small,
                             dim3((uint32_t)grid_for((int64_t)srows.size() * 64, 4096)),
                             dim3(kBlock), 0, ctx->stream, M);
        if (!brows.empty()) {
          if (pool_total > 0) {
            hipLaunchKernelGGL(k_fill32, dim3((uint32_t)grid_for((int64_t)pool_total)),
                               dim3(kBlock), 0, ctx->stream, (int64_t)pool_total, -1,
                               d_pkeys);
            MGX_HIP_TRY(hipMemsetAsync(d_pvals, 0, pool_total * 8, ctx->stream));
          }
          hipLaunchKernelGGL(
              k_move_big,
              dim3((uint32_t)(brows.size() < 4096 ? brows.size() : 4096)), dim3(kBlock),
              0, ctx->stream, M);
        }
        MGX_HIP_TRY(hipMemsetAsync(d_cnt, 0, 8, ctx->stream));
        hipLaunchKernelGGL(k_move_apply, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                           ctx->stream, nv, parity, d_target, d_comm, d_csize, d_cnt);
        unsigned long long moved = 0;
        MGX_HIP_TRY(hipMemcpyAsync(&moved, d_cnt, 8, hipMemcpyDeviceToHost, ctx->stream));
        MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
        moved_total += moved;
      }
      if (moved_total == 0) break;
      // Red-black Jacobi can park in a tiny ping-pong cycle (measured: a
      // 14-node oscillation at nv=1M held the loop at the 200-sweep cap,
      // ~20 s/level). Once moves are both negligible (<0.1% of nv) and no
      // longer decreasing, the sweep is at its fixed point modulo the
      // cycle — stop. Small graphs converge to moved==0 well before
      // sweep 10, so golden trajectories are unchanged.
      if (sweep >= 10 && moved_total * 1000 < (unsigned long long)nv &&
          moved_total >= prev_moved)
        break;
      prev_moved = moved_total;
      if (leiden_trace() && (sweep & 15) == 0)
        MGX_LEIDEN_LOG("  sweep %d moved=%llu", sweep, moved_total);
    }
    MGX_LEIDEN_LOG("  moves done (%d sweeps)", sweeps_done);

    // singleton / empties accounting
    unsigned long long nonsingleton = 0, nonempty = 0;
    {
      unsigned long long *d_t = nullptr;
      MGX_HIP_TRY(mgx_hip_malloc(&d_t, 16));
      MGX_HIP_TRY(hipMemsetAsync(d_t, 0, 16, ctx->stream));
      hipLaunchKernelGGL(k_count_nonsingleton, dim3((uint32_t)grid_for(nv)), dim3(kBlock),
                         0, ctx->stream, nv, d_csize, d_t, d_t + 1);
      unsigned long long h_t[2];
      MGX_HIP_TRY(hipMemcpyAsync(h_t, d_t, 16, hipMemcpyDeviceToHost, ctx->stream));
      MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
      nonsingleton = h_t[0];
      nonempty = h_t[1];
      (void)hipFree(d_t);
    }
    const unsigned long long move_empties =
        ((unsigned long long)nv - nonempty) - empties_before;
    done = (nonsingleton == 0);  // AllSingletonCommunities (:536)

    std::vector<int32_t> comm_h(nv);
    MGX_HIP_TRY(hipMemcpyAsync(comm_h.data(), d_comm, nv * 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));

    if (!done) {
      // ---- refinement --------------------------------------------------
      // members sorted by community (host sort; nv is the LEVEL size)
      std::vector<int32_t> members(nv);
      for (int64_t v = 0; v < nv; ++v) members[v] = (int32_t)v;
      std::sort(members.begin(), members.end(), [&](int32_t a, int32_t b) {
        if (comm_h[a] != comm_h[b]) return comm_h[a] < comm_h[b];
        return a < b;
      });
      std::vector<uint32_t> mem_off(nv + 1, 0);  // community CSR (ids 0..nv-1)
      {
        std::vector<uint32_t> cnt(nv, 0);
        for (auto v : members) cnt[comm_h[v]]++;
        for (int64_t c2 = 0; c2 < nv; ++c2) mem_off[c2 + 1] = mem_off[c2] + cnt[c2];
      }
      std::vector<int32_t> locidx(nv);
      for (int64_t i = 0; i < nv; ++i) locidx[members[i]] = 0;  // filled in kernel
      int32_t *d_members = nullptr, *d_locidx = nullptr, *d_rcomm = nullptr;
      uint32_t *d_memoff = nullptr, *d_rsize = nullptr;
      double *d_extw = nullptr, *d_edgew = nullptr, *d_prob = nullptr;
      int32_t *d_nbc = nullptr, *d_cslot = nullptr;
      MGX_HIP_TRY(mgx_hip_malloc(&d_members, nv * 4));
      MGX_HIP_TRY(mgx_hip_malloc(&d_locidx, nv * 4));
      MGX_HIP_TRY(mgx_hip_malloc(&d_rcomm, nv * 4));
      MGX_HIP_TRY(mgx_hip_malloc(&d_memoff, (nv + 1) * 4));
      MGX_HIP_TRY(mgx_hip_malloc(&d_rsize, nv * 4));
      MGX_HIP_TRY(mgx_hip_malloc(&d_extw, nv * 8));
      MGX_HIP_TRY(mgx_hip_malloc(&d_edgew, nv * 8));
      MGX_HIP_TRY(mgx_hip_malloc(&d_prob, nv * 8));
      MGX_HIP_TRY(mgx_hip_malloc(&d_nbc, nv * 4));
      MGX_HIP_TRY(mgx_hip_malloc(&d_cslot, nv * 4));
      MGX_HIP_TRY(hipMemcpyAsync(d_members, members.data(), nv * 4,
                                 hipMemcpyHostToDevice, ctx->stream));
      MGX_HIP_TRY(hipMemcpyAsync(d_memoff, mem_off.data(), (nv + 1) * 4,
                                 hipMemcpyHostToDevice, ctx->stream));
      // rcomm starts singleton-local: filled per subset inside k_refine
      hipLaunchKernelGGL(k_fill32, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                         ctx->stream, nv, 0, d_rcomm);
      MGX_HIP_TRY(hipMemsetAsync(d_cnt, 0, 8, ctx->stream));
      RefineArgs R;
      R.row_ptr = L.row_ptr;
      R.col = L.col;
      R.w = L.w;
      R.node_w = L.node_w;
      R.comm = d_comm;
      R.members = d_members;
      R.mem_off = d_memoff;
      R.n_comm = nv;
      R.loc_idx = d_locidx;
      R.rcomm_loc = d_rcomm;
      R.ext_w = d_extw;
      R.edge_w = d_edgew;
      R.nb_comms = d_nbc;
      R.nb_slot = d_cslot;
      R.prob = d_prob;
      R.rsize = d_rsize;
      R.gamma = gamma_n;
      R.theta = theta;
      R.resolution = resolution;
      R.seed = mgx_hash64(ms, (uint64_t)level + 1);
      R.n_merged = d_cnt;
      MGX_LEIDEN_LOG("  refine launch n_comm=%lld", (long long)nv);
      hipLaunchKernelGGL(k_refine, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                         ctx->stream, R);
      unsigned long long merged = 0;
      MGX_HIP_TRY(hipMemcpyAsync(&merged, d_cnt, 8, hipMemcpyDeviceToHost, ctx->stream));
      std::vector<int32_t> rcomm_h(nv);
      MGX_HIP_TRY(hipMemcpyAsync(rcomm_h.data(), d_rcomm, nv * 4, hipMemcpyDeviceToHost,
                                 ctx->stream));
      MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
      (void)hipFree(d_members);
      (void)hipFree(d_locidx);
      (void)hipFree(d_memoff);
      (void)hipFree(d_rsize);
      (void)hipFree(d_extw);
      (void)hipFree(d_edgew);
      (void)hipFree(d_prob);
      (void)hipFree(d_nbc);
      (void)hipFree(d_cslot);
      (void)hipFree(d_rcomm);

      // global refined community id per node: the member slot of its local
      // refined community (== another node's global member position)
      std::vector<int32_t> refined(nv);
      for (int64_t i = 0; i < nv; ++i) {
        const int32_t v = members[i];
        // rcomm_h[v] is a LOCAL index within v's subset for subsets the
        // kernel touched; for singleton subsets it stayed 0 == local self
        const int32_t c2 = comm_h[v];
        refined[v] = (int32_t)mem_off[c2] + rcomm_h[v];
      }
      // refined ids are member positions; map to node ids for stability:
      for (int64_t v = 0; v < nv; ++v) refined[v] = members[refined[v]];

      const unsigned long long refined_empties = merged;
      bool use_move_partition = false;
      if (refined_empties == (unsigned long long)nv - 1) {
        done = true;  // everything merged into one (:542-545; isolated-node
                      // term is dead code — Graph(size) keeps Size()==N)
      } else if (refined_empties == 0) {
        // empty_communities == partitions.communities.size() - 1 (:549-552):
        // groups-at-level-start minus one all emptied during the moves
        const unsigned long long groups_before =
            (unsigned long long)nv - empties_before;
        if (groups_before >= 1 && move_empties == groups_before - 1) done = true;
        use_move_partition = true;  // refined := partitions (:553-556)
      }
      if (!done) {
        const std::vector<int32_t> &agg_refined = use_move_partition ? comm_h : refined;
        const std::vector<int32_t> &agg_original = comm_h;
        MGX_LEIDEN_LOG("  aggregate start");
        // ---- aggregate (host; level sizes shrink fast) ------------------
        // compact refined community ids
        std::unordered_map<int32_t, int32_t> remap;
        std::vector<int64_t> parent(nv);
        std::vector<int32_t> rep_order;  // refined id per new community
        for (int64_t i = 0; i < nv; ++i) {
          const int32_t v = members[i];  // ascending-community member order
          const int32_t rc = agg_refined[v];
          auto it = remap.find(rc);
          int32_t nid;
          if (it == remap.end()) {
            nid = (int32_t)remap.size();
            remap.emplace(rc, nid);
            rep_order.push_back(rc);
          } else {
            nid = it->second;
          }
          parent[v] = nid;
        }
        const int64_t n_new = (int64_t)remap.size();
        dendro.push_back(parent);  // level-entity -> next-level id

        // aggregated simple graph: first edge per unordered new pair
        std::vector<uint32_t> rp_new(n_new + 1, 0);
        std::vector<std::vector<std::pair<int32_t, double>>> adj(n_new);
        std::vector<double> nw_new(n_new, 0.0);
        {
          std::vector<double> w_l(L.ne2);
          std::vector<int32_t> col_l(L.ne2);
          MGX_HIP_TRY(hipMemcpyAsync(col_l.data(), L.col, L.ne2 * 4,
                                     hipMemcpyDeviceToHost, ctx->stream));
          MGX_HIP_TRY(hipMemcpyAsync(w_l.data(), L.w, L.ne2 * 8, hipMemcpyDeviceToHost,
                                     ctx->stream));
          MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
          std::unordered_set<uint64_t> seen;
          seen.reserve((size_t)(L.ne2 / 2 + 16));
          for (int64_t v = 0; v < nv; ++v) {
            const int32_t cu = (int32_t)parent[v];
            for (uint32_t j = rp_l[v]; j < rp_l[v + 1]; ++j) {
              const int32_t cv2 = (int32_t)parent[col_l[j]];
              if (cu == cv2) continue;
              const uint64_t key = cu < cv2
                                       ? ((uint64_t)(uint32_t)cu << 32) | (uint32_t)cv2
                                       : ((uint64_t)(uint32_t)cv2 << 32) | (uint32_t)cu;
              if (!seen.insert(key).second) continue;
              adj[cu].emplace_back(cv2, w_l[j]);
              adj[cv2].emplace_back(cu, w_l[j]);
              nw_new[cu] += w_l[j];
              nw_new[cv2] += w_l[j];
            }
          }
        }
        // partitions for the next level: group new ids by ORIGINAL
        // partition of their representative (:424-443)
        std::vector<int32_t> comm_new(n_new);
        {
          std::unordered_map<int32_t, int32_t> orig2new;
          for (int64_t nid = 0; nid < n_new; ++nid) {
            const int32_t rep = rep_order[nid];  // a node id of the level
            const int32_t oc = agg_original[rep];
            auto it = orig2new.find(oc);
            int32_t pc;
            if (it == orig2new.end()) {
              pc = (int32_t)orig2new.size();
              orig2new.emplace(oc, pc);
            } else {
              pc = it->second;
            }
            comm_new[nid] = pc;
          }
        }

        // upload the new level
        free_level(&L);
        L.nv = n_new;
        std::vector<uint32_t> rp_up(n_new + 1, 0);
        std::vector<int32_t> col_up;
        std::vector<double> w_up;
        for (int64_t i2 = 0; i2 < n_new; ++i2) {
          for (auto &[c2, wv] : adj[i2]) {
            col_up.push_back(c2);
            w_up.push_back(wv);
          }
          rp_up[i2 + 1] = (uint32_t)col_up.size();
        }
        L.ne2 = (int64_t)col_up.size();
        MGX_HIP_TRY(mgx_hip_malloc(&L.row_ptr, (n_new + 1) * 4));
        MGX_HIP_TRY(mgx_hip_malloc(&L.col, (L.ne2 ? L.ne2 : 1) * 4));
        MGX_HIP_TRY(mgx_hip_malloc(&L.w, (L.ne2 ? L.ne2 : 1) * 8));
        MGX_HIP_TRY(mgx_hip_malloc(&L.node_w, n_new * 8));
        MGX_HIP_TRY(hipMemcpyAsync(L.row_ptr, rp_up.data(), (n_new + 1) * 4,
                                   hipMemcpyHostToDevice, ctx->stream));
        if (L.ne2) {
          MGX_HIP_TRY(hipMemcpyAsync(L.col, col_up.data(), L.ne2 * 4,
                                     hipMemcpyHostToDevice, ctx->stream));
          MGX_HIP_TRY(hipMemcpyAsync(L.w, w_up.data(), L.ne2 * 8, hipMemcpyHostToDevice,
                                     ctx->stream));
        }
        MGX_HIP_TRY(hipMemcpyAsync(L.node_w, nw_new.data(), n_new * 8,
                                   hipMemcpyHostToDevice, ctx->stream));
        (void)hipFree(d_comm);
        (void)hipFree(d_csize);
        MGX_HIP_TRY(mgx_hip_malloc(&d_comm, n_new * 4));
        MGX_HIP_TRY(mgx_hip_malloc(&d_csize, n_new * 4));
        MGX_HIP_TRY(hipMemcpyAsync(d_comm, comm_new.data(), n_new * 4,
                                   hipMemcpyHostToDevice, ctx->stream));
        std::vector<uint32_t> cs_new((size_t)n_new, 0);
        for (auto c2 : comm_new) cs_new[c2]++;
        // csize indexed by community id; pad to n_new
        std::vector<uint32_t> cs_arr((size_t)n_new, 0);
        for (int64_t nid = 0; nid < n_new; ++nid) cs_arr[comm_new[nid]]++;
        MGX_HIP_TRY(hipMemcpyAsync(d_csize, cs_arr.data(), n_new * 4,
                                   hipMemcpyHostToDevice, ctx->stream));
        MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
        ++level;
      }
    }
    (void)hipFree(d_srows);
    (void)hipFree(d_brows);
    (void)hipFree(d_target);
    (void)hipFree(d_poff);
    (void)hipFree(d_pkeys);
    (void)hipFree(d_pvals);
    (void)hipFree(d_cnt);
    if (iters >= max_iterations) done = true;
  }
  (void)hipFree(d_comm);
  (void)hipFree(d_csize);
  free_level(&L);
  if (st != MGX_OK) return st;

  // compose per-node hierarchies from the dendrogram levels
  for (int64_t v = 0; v < V; ++v) {
    int64_t cur = v;
    int64_t Lc = 0;
    for (auto &lvl : dendro) {
      cur = lvl[cur];
      if (Lc < cap) out_hier[v * cap + Lc] = cur;
      ++Lc;
    }
    out_levels[v] = Lc;
  }
  return MGX_OK;
}
