// Online (dynamic) PageRank on gfx950 — replaces pagerank_online_alg
// (reference query_modules/pagerank_module/algorithm_online/pagerank.cpp):
//
//   SetPagerank   (:253-278): R random walks from every node; a walk appends
//                 random out-neighbours until a stop draw < epsilon or a
//                 dangling node; rank[v] = visits(v) / total visits (the
//                 ((n*R)/eps) scaling at :92 cancels in NormalizeRank :70-76).
//   UpdatePagerank(:292-315): per deleted/created edge (from,to), every walk
//                 containing `from` is truncated after its FIRST occurrence
//                 (:150-157, :204-216) and regrown from `from` with eps/2
//                 (:160-161, :225-226); created vertices get R fresh walks
//                 (:172-188); deleted vertices lose their counter entries
//                 (:236-239).
//   GetPagerank   (:280-290): recompute from state; inconsistent when a graph
//                 node has no walk state (:241-250).
//
// MI355X-native state layout (DESIGN.md "Statistical-parity bar"): one device
// entry pool of (walk u32, pos u32, node-slot i32) triples — the walks
// themselves are the ground truth, exactly as the reference's walks vector
// is — plus per-walk start/generation/liveness. Truncation marks entries
// dead (node = -1); counters are recomputed by a histogram over live
// entries, which removes the reference's decrement bookkeeping (and its
// unsigned-underflow bug on revived vertices, which we do NOT replicate).
// Walk generation/regrowth is one device thread per walk with counter-based
// splitmix64 streams keyed (walk, generation, step), so a capacity-overflow
// retry regenerates identical walks.
//
// Known, documented divergences from the reference (statistical bar applies):
//  - RNG: seeded splitmix64 streams instead of std::random_device+minstd;
//    neighbour pick is h % deg (modulo bias ~2^-32).
//  - Edge updates within one call are processed in the reference's order,
//    each fully parallel over affected walks.
//  - Walk length is capped at 1<<20 steps (P(hit) < eps*(1-eps)^1e6 ~ 0).

#include <unordered_map>
#include <vector>

#include "../../include/mgx_graphgen.h"
#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;
constexpr uint32_t kMaxSteps = 1u << 20;

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

struct PrOnlineState {
  // host slot map: memgraph id -> slot (stable across graph changes)
  std::unordered_map<int64_t, int32_t> mg2slot;
  std::vector<int64_t> slot2mg;
  std::vector<uint8_t> slot_alive_h;

  // device
  uint32_t *e_walk = nullptr;
  uint32_t *e_pos = nullptr;
  int32_t *e_node = nullptr;  // slot id; -1 = dead entry
  int64_t pool_cap = 0;
  unsigned long long *d_cursor = nullptr;  // [0]=append cursor, [1]=overflow
  int64_t pool_used = 0;                   // host mirror (valid after sync)

  int32_t *w_start = nullptr;  // slot
  uint32_t *w_gen = nullptr;   // regeneration counter (RNG stream)
  uint8_t *w_dead = nullptr;
  int64_t n_walks = 0, walks_cap = 0;

  uint8_t *d_slot_alive = nullptr;
  int64_t slots_cap_dev = 0;

  uint64_t seed_mixed = 0;
  int64_t R = 10;
  double eps = 0.2;
  bool initialized = false;
};

PrOnlineState g_st;

// ---- kernels -------------------------------------------------------------

struct GenItem {
  uint32_t walk;
  int32_t start_slot;
  uint32_t start_pos;   // position of the first APPENDED entry
  uint32_t gen;         // RNG stream generation
  uint8_t include_start;  // 1: append (walk, 0, start) first (new walks)
};

struct GenArgs {
  const GenItem *items;
  int64_t n_items;
  const uint32_t *row_ptr;  // out-CSR, dense space
  const int32_t *col;
  const int32_t *slot2dense;  // slot -> dense (current graph), -1 if absent
  const int32_t *dense2slot;  // dense -> slot
  uint32_t *e_walk;
  uint32_t *e_pos;
  int32_t *e_node;
  unsigned long long *cursor;  // [0] append, [1] overflow flag
  uint64_t pool_cap;
  uint64_t seed;  // mixed
  uint64_t eps_bits;  // stop threshold as u64 (draw < eps_bits stops)
};

__device__ inline bool pool_append(const GenArgs &A, uint32_t w, uint32_t pos,
                                   int32_t slot) {
  unsigned long long at = atomicAdd(A.cursor, 1ull);
  if (at >= A.pool_cap) {
    atomicExch(A.cursor + 1, 1ull);
    return false;
  }
  A.e_walk[at] = w;
  A.e_pos[at] = pos;
  A.e_node[at] = slot;
  return true;
}

// One thread per walk item: replicates CreateRoute (pagerank.cpp:110-134).
__global__ void k_walk_gen(GenArgs A) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < A.n_items;
       i += (int64_t)gridDim.x * blockDim.x) {
    const GenItem it = A.items[i];
    if (it.include_start) {
      if (!pool_append(A, it.walk, 0, it.start_slot)) return;
    }
    // per-(walk, gen) RNG stream
    const uint64_t stream =
        mgx_hash64(A.seed, ((uint64_t)it.walk << 24) ^ (uint64_t)it.gen);
    int32_t cur_slot = it.start_slot;
    uint32_t pos = it.start_pos;
    for (uint32_t step = 0; step < kMaxSteps; ++step) {
      const int32_t dense = A.slot2dense[cur_slot];
      if (dense < 0) break;  // node no longer in the graph (defensive)
      const uint32_t s = A.row_ptr[dense], e = A.row_ptr[dense + 1];
      const uint32_t deg = e - s;
      if (deg == 0) break;  // dangling: walk ends (pagerank.cpp:114-115)
      const uint64_t h1 = mgx_hash64(stream, 2ull * step);
      const int32_t nb_dense = A.col[s + (uint32_t)(h1 % deg)];
      const int32_t nb_slot = A.dense2slot[nb_dense];
      if (!pool_append(A, it.walk, pos, nb_slot)) return;
      ++pos;
      const uint64_t h2 = mgx_hash64(stream, 2ull * step + 1);
      if (h2 < A.eps_bits) break;  // stop draw (pagerank.cpp:128-130)
      cur_slot = nb_slot;
    }
  }
}

// First-occurrence position of `from` per walk (walks_table lookup +
// std::find, pagerank.cpp:146-150).
__global__ void k_find_affected(int64_t used, const uint32_t *e_walk, const uint32_t *e_pos,
                                const int32_t *e_node, const uint8_t *w_dead,
                                int32_t from_slot, uint32_t *minpos) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < used;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (e_node[i] == from_slot && !w_dead[e_walk[i]])
      atomicMin(&minpos[e_walk[i]], e_pos[i]);
  }
}

__global__ void k_collect_affected(int64_t n_walks, const uint32_t *minpos,
                                   uint32_t *out_walks, unsigned long long *n_out) {
  for (int64_t w = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; w < n_walks;
       w += (int64_t)gridDim.x * blockDim.x) {
    if (minpos[w] != 0xFFFFFFFFu) {
      unsigned long long at = atomicAdd(n_out, 1ull);
      out_walks[at] = (uint32_t)w;
    }
  }
}

// walk.erase(first_occurrence+1, end) (pagerank.cpp:157/216)
__global__ void k_truncate(int64_t used, const uint32_t *e_walk, const uint32_t *e_pos,
                           int32_t *e_node, const uint32_t *minpos) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < used;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t mp = minpos[e_walk[i]];
    if (mp != 0xFFFFFFFFu && e_pos[i] > mp) e_node[i] = -1;
  }
}

__global__ void k_make_regrow_items(int64_t n_aff, const uint32_t *aff_walks,
                                    const uint32_t *minpos, int32_t from_slot,
                                    uint32_t *w_gen, GenItem *items) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_aff;
       i += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t w = aff_walks[i];
    items[i].walk = w;
    items[i].start_slot = from_slot;
    items[i].start_pos = minpos[w] + 1;
    items[i].gen = ++w_gen[w];
    items[i].include_start = 0;
  }
}

__global__ void k_mark_dead_walks(int64_t n_walks, const int32_t *w_start,
                                  int32_t slot, uint8_t *w_dead) {
  for (int64_t w = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; w < n_walks;
       w += (int64_t)gridDim.x * blockDim.x) {
    if (w_start[w] == slot) w_dead[w] = 1;
  }
}

__global__ void k_histogram(int64_t used, const uint32_t *e_walk, const int32_t *e_node,
                            const uint8_t *w_dead, const uint8_t *slot_alive,
                            uint32_t *counter) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < used;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t n = e_node[i];
    if (n >= 0 && !w_dead[e_walk[i]] && slot_alive[n]) atomicAdd(&counter[n], 1u);
  }
}

__global__ void k_count_live(int64_t used, const int32_t *e_node, const uint32_t *e_walk,
                             const uint8_t *w_dead, unsigned long long *out) {
  unsigned long long acc = 0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < used;
       i += (int64_t)gridDim.x * blockDim.x)
    if (e_node[i] >= 0 && !w_dead[e_walk[i]]) ++acc;
  atomicAdd(out, acc);
}

__global__ void k_sum_u32(int64_t n, const uint32_t *x, unsigned long long *out) {
  unsigned long long acc = 0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    acc += x[i];
  __shared__ unsigned long long red[kBlock / 64];
  for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    unsigned long long s = 0;
    for (int i = 0; i < kBlock / 64; ++i) s += red[i];
    atomicAdd(out, s);
  }
}

__global__ void k_rank_out(int64_t V, const int32_t *dense2slot, const uint32_t *counter,
                           const unsigned long long *sum, double *rank) {
  const double inv = *sum > 0 ? 1.0 / (double)*sum : 0.0;
  for (int64_t v = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; v < V;
       v += (int64_t)gridDim.x * blockDim.x) {
    const int32_t s = dense2slot[v];
    rank[v] = s >= 0 ? (double)counter[s] * inv : 0.0;
  }
}

// ---- host helpers --------------------------------------------------------

mgx_status ensure_pool(mgx_context *ctx, int64_t need_cap) {
  if (need_cap <= g_st.pool_cap) return MGX_OK;
  int64_t cap = g_st.pool_cap > 0 ? g_st.pool_cap : (1 << 20);
  while (cap < need_cap) cap *= 2;
  uint32_t *nw = nullptr, *np = nullptr;
  int32_t *nn = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&nw, cap * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&np, cap * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&nn, cap * 4));
  if (g_st.pool_used > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(nw, g_st.e_walk, g_st.pool_used * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(np, g_st.e_pos, g_st.pool_used * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(nn, g_st.e_node, g_st.pool_used * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }
  if (g_st.e_walk) (void)hipFree(g_st.e_walk);
  if (g_st.e_pos) (void)hipFree(g_st.e_pos);
  if (g_st.e_node) (void)hipFree(g_st.e_node);
  g_st.e_walk = nw;
  g_st.e_pos = np;
  g_st.e_node = nn;
  g_st.pool_cap = cap;
  return MGX_OK;
}

mgx_status ensure_walks(mgx_context *ctx, int64_t need) {
  if (need <= g_st.walks_cap) return MGX_OK;
  int64_t cap = g_st.walks_cap > 0 ? g_st.walks_cap : (1 << 16);
  while (cap < need) cap *= 2;
  int32_t *ns = nullptr;
  uint32_t *ng = nullptr;
  uint8_t *nd = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&ns, cap * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&ng, cap * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&nd, cap));
  MGX_HIP_TRY(hipMemsetAsync(ng, 0, cap * 4, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(nd, 0, cap, ctx->stream));
  if (g_st.n_walks > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(ns, g_st.w_start, g_st.n_walks * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(ng, g_st.w_gen, g_st.n_walks * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(nd, g_st.w_dead, g_st.n_walks,
                               hipMemcpyDeviceToDevice, ctx->stream));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  if (g_st.w_start) (void)hipFree(g_st.w_start);
  if (g_st.w_gen) (void)hipFree(g_st.w_gen);
  if (g_st.w_dead) (void)hipFree(g_st.w_dead);
  g_st.w_start = ns;
  g_st.w_gen = ng;
  g_st.w_dead = nd;
  g_st.walks_cap = cap;
  return MGX_OK;
}

mgx_status sync_slot_alive(mgx_context *ctx) {
  const int64_t n = (int64_t)g_st.slot_alive_h.size();
  if (n > g_st.slots_cap_dev) {
    if (g_st.d_slot_alive) (void)hipFree(g_st.d_slot_alive);
    int64_t cap = 64;
    while (cap < n) cap *= 2;
    MGX_HIP_TRY(mgx_hip_malloc(&g_st.d_slot_alive, cap));
    g_st.slots_cap_dev = cap;
  }
  if (n > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(g_st.d_slot_alive, g_st.slot_alive_h.data(), n,
                               hipMemcpyHostToDevice, ctx->stream));
  }
  return MGX_OK;
}

void free_state() {
  if (g_st.e_walk) (void)hipFree(g_st.e_walk);
  if (g_st.e_pos) (void)hipFree(g_st.e_pos);
  if (g_st.e_node) (void)hipFree(g_st.e_node);
  if (g_st.w_start) (void)hipFree(g_st.w_start);
  if (g_st.w_gen) (void)hipFree(g_st.w_gen);
  if (g_st.w_dead) (void)hipFree(g_st.w_dead);
  if (g_st.d_cursor) (void)hipFree(g_st.d_cursor);
  if (g_st.d_slot_alive) (void)hipFree(g_st.d_slot_alive);
  g_st = PrOnlineState{};
}

// Slot of mg id, creating if absent.
int32_t slot_of(int64_t mg_id) {
  auto it = g_st.mg2slot.find(mg_id);
  if (it != g_st.mg2slot.end()) return it->second;
  const int32_t s = (int32_t)g_st.slot2mg.size();
  g_st.mg2slot.emplace(mg_id, s);
  g_st.slot2mg.push_back(mg_id);
  g_st.slot_alive_h.push_back(1);
  return s;
}

struct Maps {
  int32_t *slot2dense = nullptr;  // device
  int32_t *dense2slot = nullptr;  // device
  int64_t V = 0;
  mgx_context *ctx = nullptr;
  ~Maps() {
    if (slot2dense) (void)hipFree(slot2dense);
    if (dense2slot) (void)hipFree(dense2slot);
  }
};

// Build device slot<->dense maps for the current scan; creates slots for
// any unseen mg ids when `create` (set path), else leaves them unmapped.
mgx_status build_maps(mgx_context *ctx, const int64_t *dense_to_mg, int64_t V, bool create,
                      Maps *m, bool *all_known) {
  m->ctx = ctx;
  m->V = V;
  std::vector<int32_t> d2s(V > 0 ? V : 1, -1);
  bool known = true;
  for (int64_t v = 0; v < V; ++v) {
    if (create) {
      d2s[v] = slot_of(dense_to_mg[v]);
    } else {
      auto it = g_st.mg2slot.find(dense_to_mg[v]);
      if (it == g_st.mg2slot.end() || !g_st.slot_alive_h[it->second]) {
        known = false;
        d2s[v] = -1;
      } else {
        d2s[v] = it->second;
      }
    }
  }
  const int64_t n_slots = (int64_t)g_st.slot2mg.size();
  std::vector<int32_t> s2d(n_slots > 0 ? n_slots : 1, -1);
  for (int64_t v = 0; v < V; ++v)
    if (d2s[v] >= 0) s2d[d2s[v]] = (int32_t)v;
  MGX_HIP_TRY(mgx_hip_malloc(&m->dense2slot, (V > 0 ? V : 1) * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&m->slot2dense, (n_slots > 0 ? n_slots : 1) * 4));
  MGX_HIP_TRY(hipMemcpyAsync(m->dense2slot, d2s.data(), (V > 0 ? V : 1) * 4,
                             hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipMemcpyAsync(m->slot2dense, s2d.data(), (n_slots > 0 ? n_slots : 1) * 4,
                             hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  if (all_known) *all_known = known;
  return MGX_OK;
}

// Run a generation batch with the grow-and-retry capacity protocol: the
// cursor snapshot is restored and the batch re-runs (same RNG streams) if
// the pool overflows.
mgx_status run_gen(mgx_context *ctx, const Maps &m, mgx_graph *g,
                   const GenItem *d_items, int64_t n_items, double eps,
                   int64_t expected_append) {
  if (n_items == 0) return MGX_OK;
  if (!g_st.d_cursor) {
    MGX_HIP_TRY(mgx_hip_malloc(&g_st.d_cursor, 16));
  }
  uint64_t eps_bits;
  if (eps >= 1.0) {
    eps_bits = ~0ull;
  } else if (eps <= 0.0) {
    eps_bits = 0;
  } else {
    eps_bits = (uint64_t)(eps * 18446744073709551616.0);
  }
  GenArgs A;
  A.items = d_items;
  A.n_items = n_items;
  A.row_ptr = g ? g->out_row_ptr : nullptr;  // never dereferenced: with no
  A.col = g ? g->out_col : nullptr;          // graph every slot2dense is -1
  A.slot2dense = m.slot2dense;
  A.dense2slot = m.dense2slot;
  A.seed = g_st.seed_mixed;
  A.eps_bits = eps_bits;
  const unsigned long long snapshot = (unsigned long long)g_st.pool_used;
  for (int attempt = 0; attempt < 40; ++attempt) {
    MGX_TRY(ensure_pool(ctx, g_st.pool_used + expected_append));
    A.e_walk = g_st.e_walk;
    A.e_pos = g_st.e_pos;
    A.e_node = g_st.e_node;
    A.cursor = g_st.d_cursor;
    A.pool_cap = (uint64_t)g_st.pool_cap;
    unsigned long long init[2] = {snapshot, 0};
    MGX_HIP_TRY(hipMemcpyAsync(g_st.d_cursor, init, 16, hipMemcpyHostToDevice,
                               ctx->stream));
    hipLaunchKernelGGL(k_walk_gen, dim3((uint32_t)grid_for(n_items)), dim3(kBlock), 0,
                       ctx->stream, A);
    unsigned long long out[2] = {0, 0};
    MGX_HIP_TRY(hipMemcpyAsync(out, g_st.d_cursor, 16, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (!out[1] && out[0] <= (unsigned long long)g_st.pool_cap) {
      g_st.pool_used = (int64_t)out[0];
      return MGX_OK;
    }
    // overflow: grow to the observed demand and retry the whole batch
    expected_append = (int64_t)(out[0] - snapshot) + (int64_t)(out[0] - snapshot) / 2 + 1024;
  }
  mgx_set_error("pronline: walk pool growth did not converge");
  return MGX_ERR_OUT_OF_MEMORY;
}

mgx_status compute_rank(mgx_context *ctx, const Maps &m, double *out_rank) {
  const int64_t n_slots = (int64_t)g_st.slot2mg.size();
  MGX_TRY(sync_slot_alive(ctx));
  uint32_t *counter = nullptr;
  unsigned long long *d_sum = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&counter, (n_slots > 0 ? n_slots : 1) * 4));
  MGX_TRY(ctx->alloc_async((void **)&d_sum, 8));
  MGX_HIP_TRY(hipMemsetAsync(counter, 0, (n_slots > 0 ? n_slots : 1) * 4, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(d_sum, 0, 8, ctx->stream));
  if (g_st.pool_used > 0)
    hipLaunchKernelGGL(k_histogram, dim3((uint32_t)grid_for(g_st.pool_used)), dim3(kBlock),
                       0, ctx->stream, g_st.pool_used, g_st.e_walk, g_st.e_node,
                       g_st.w_dead, g_st.d_slot_alive, counter);
  if (n_slots > 0)
    hipLaunchKernelGGL(k_sum_u32, dim3((uint32_t)grid_for(n_slots)), dim3(kBlock), 0,
                       ctx->stream, n_slots, counter, d_sum);
  if (out_rank && m.V > 0) {
    double *d_rank = nullptr;
    MGX_TRY(ctx->alloc_async((void **)&d_rank, m.V * 8));
    hipLaunchKernelGGL(k_rank_out, dim3((uint32_t)grid_for(m.V)), dim3(kBlock), 0,
                       ctx->stream, m.V, m.dense2slot, counter, d_sum, d_rank);
    MGX_HIP_TRY(hipMemcpyAsync(out_rank, d_rank, m.V * 8, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    MGX_TRY(ctx->free_async(d_rank));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_TRY(ctx->free_async(counter));
  MGX_TRY(ctx->free_async(d_sum));
  return MGX_OK;
}

// R fresh walks from each listed slot (SetPagerank :260-275 /
// UpdateCreate(vertex) :172-188).
mgx_status spawn_walks(mgx_context *ctx, const Maps &m, mgx_graph *g,
                       const std::vector<int32_t> &starts) {
  const int64_t n_new = (int64_t)starts.size() * g_st.R;
  if (n_new == 0) return MGX_OK;
  MGX_TRY(ensure_walks(ctx, g_st.n_walks + n_new));
  std::vector<GenItem> items(n_new);
  std::vector<int32_t> wstart(n_new);
  for (int64_t i = 0; i < (int64_t)starts.size(); ++i) {
    for (int64_t r = 0; r < g_st.R; ++r) {
      const int64_t k = i * g_st.R + r;
      items[k].walk = (uint32_t)(g_st.n_walks + k);
      items[k].start_slot = starts[i];
      items[k].start_pos = 1;
      items[k].gen = 0;
      items[k].include_start = 1;
      wstart[k] = starts[i];
    }
  }
  MGX_HIP_TRY(hipMemcpyAsync(g_st.w_start + g_st.n_walks, wstart.data(), n_new * 4,
                             hipMemcpyHostToDevice, ctx->stream));
  GenItem *d_items = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&d_items, n_new * sizeof(GenItem)));
  MGX_HIP_TRY(hipMemcpyAsync(d_items, items.data(), n_new * sizeof(GenItem),
                             hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  g_st.n_walks += n_new;
  const int64_t expect = n_new * (int64_t)(2.0 / (g_st.eps > 0.01 ? g_st.eps : 0.01) + 2);
  mgx_status s = run_gen(ctx, m, g, d_items, n_new, g_st.eps, expect);
  (void)ctx->free_async(d_items);
  return s;
}

// Truncate-and-regrow for one changed edge's `from` endpoint
// (UpdateCreate(edge) :143-163 / UpdateDelete(edge) :197-228).
mgx_status rewire_from(mgx_context *ctx, const Maps &m, mgx_graph *g, int32_t from_slot,
                       bool from_exists) {
  if (g_st.n_walks == 0 || g_st.pool_used == 0) return MGX_OK;
  uint32_t *minpos = nullptr;
  uint32_t *aff = nullptr;
  unsigned long long *n_aff_d = nullptr;
  MGX_TRY(ctx->alloc_async((void **)&minpos, g_st.n_walks * 4));
  MGX_TRY(ctx->alloc_async((void **)&aff, g_st.n_walks * 4));
  MGX_TRY(ctx->alloc_async((void **)&n_aff_d, 8));
  MGX_HIP_TRY(hipMemsetAsync(minpos, 0xFF, g_st.n_walks * 4, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(n_aff_d, 0, 8, ctx->stream));
  hipLaunchKernelGGL(k_find_affected, dim3((uint32_t)grid_for(g_st.pool_used)),
                     dim3(kBlock), 0, ctx->stream, g_st.pool_used, g_st.e_walk,
                     g_st.e_pos, g_st.e_node, g_st.w_dead, from_slot, minpos);
  hipLaunchKernelGGL(k_truncate, dim3((uint32_t)grid_for(g_st.pool_used)), dim3(kBlock), 0,
                     ctx->stream, g_st.pool_used, g_st.e_walk, g_st.e_pos, g_st.e_node,
                     minpos);
  hipLaunchKernelGGL(k_collect_affected, dim3((uint32_t)grid_for(g_st.n_walks)),
                     dim3(kBlock), 0, ctx->stream, g_st.n_walks, minpos, aff, n_aff_d);
  unsigned long long n_aff = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&n_aff, n_aff_d, 8, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  mgx_status s = MGX_OK;
  if (n_aff > 0 && from_exists) {
    GenItem *d_items = nullptr;
    MGX_TRY(ctx->alloc_async((void **)&d_items, n_aff * sizeof(GenItem)));
    hipLaunchKernelGGL(k_make_regrow_items, dim3((uint32_t)grid_for((int64_t)n_aff)),
                       dim3(kBlock), 0, ctx->stream, (int64_t)n_aff, aff, minpos,
                       from_slot, g_st.w_gen, d_items);
    const int64_t expect =
        (int64_t)n_aff * (int64_t)(4.0 / (g_st.eps > 0.01 ? g_st.eps : 0.01) + 2);
    s = run_gen(ctx, m, g, d_items, (int64_t)n_aff, g_st.eps / 2.0, expect);
    (void)ctx->free_async(d_items);
  }
  (void)ctx->free_async(minpos);
  (void)ctx->free_async(aff);
  (void)ctx->free_async(n_aff_d);
  return s;
}

}  // namespace

extern "C" int mgx_pronline_initialized(void) { return g_st.initialized ? 1 : 0; }

extern "C" mgx_status mgx_pronline_reset(mgx_context *ctx) {
  (void)ctx;
  free_state();
  return MGX_OK;
}

extern "C" mgx_status mgx_pronline_set(mgx_context *ctx, mgx_graph *g,
                                       const int64_t *dense_to_mg, int64_t R, double eps,
                                       uint64_t seed, double *out_rank) {
  if (g && !(g->flags & MGX_BUILD_OUT_CSR)) {
    mgx_set_error("pronline_set needs a graph built with MGX_BUILD_OUT_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  free_state();
  g_st.R = R > 0 ? R : 1;
  g_st.eps = eps;
  g_st.seed_mixed = mgx_seed_mix(seed);
  g_st.initialized = true;
  const int64_t V = g ? g->n_vertices : 0;  // null graph == empty graph
  Maps m;
  MGX_TRY(build_maps(ctx, dense_to_mg, V, /*create=*/true, &m, nullptr));
  std::vector<int32_t> starts(V);
  for (int64_t v = 0; v < V; ++v) starts[v] = (int32_t)v;  // slots == dense here
  MGX_TRY(spawn_walks(ctx, m, g, starts));
  return compute_rank(ctx, m, out_rank);
}

extern "C" mgx_status mgx_pronline_get(mgx_context *ctx, const int64_t *dense_to_mg,
                                       int64_t V, double *out_rank, int *consistent) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  Maps m;
  bool known = true;
  MGX_TRY(build_maps(ctx, dense_to_mg, V, /*create=*/false, &m, &known));
  if (consistent) *consistent = known ? 1 : 0;
  if (!known) return MGX_OK;
  return compute_rank(ctx, m, out_rank);
}

extern "C" mgx_status mgx_pronline_update(mgx_context *ctx, mgx_graph *g,
                                          const int64_t *dense_to_mg,
                                          const int64_t *created_v, int64_t n_cv,
                                          const int64_t *created_e, int64_t n_ce,
                                          const int64_t *deleted_v, int64_t n_dv,
                                          const int64_t *deleted_e, int64_t n_de,
                                          double *out_rank) {
  if (g && !(g->flags & MGX_BUILD_OUT_CSR)) {
    mgx_set_error("pronline_update needs a graph built with MGX_BUILD_OUT_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  const int64_t V = g ? g->n_vertices : 0;  // null graph == empty graph

  // UpdatePagerank order (pagerank.cpp:301-312): deleted edges, deleted
  // vertices, created vertices, created edges.
  // Slots for created vertices must exist before edge rewires reference
  // them; build the maps AFTER creating/reviving slots.
  for (int64_t i = 0; i < n_cv; ++i) {
    const int32_t s = slot_of(created_v[i]);
    g_st.slot_alive_h[s] = 1;
  }
  Maps m;
  MGX_TRY(build_maps(ctx, dense_to_mg, V, /*create=*/false, &m, nullptr));

  // Scan membership (mg id -> in current graph) for the NodeExists check
  // (pagerank.cpp:221-223): regrow only if `from` is in the post-change
  // graph.
  std::unordered_map<int64_t, int32_t> mg2dense;
  mg2dense.reserve((size_t)V * 2);
  for (int64_t v = 0; v < V; ++v) mg2dense.emplace(dense_to_mg[v], (int32_t)v);

  for (int64_t i = 0; i < n_de; ++i) {
    const int64_t from = deleted_e[2 * i];
    auto it = g_st.mg2slot.find(from);
    if (it == g_st.mg2slot.end()) continue;
    const bool exists = mg2dense.count(from) > 0;
    MGX_TRY(rewire_from(ctx, m, g, it->second, exists));
  }
  for (int64_t i = 0; i < n_dv; ++i) {
    auto it = g_st.mg2slot.find(deleted_v[i]);
    if (it == g_st.mg2slot.end()) continue;
    g_st.slot_alive_h[it->second] = 0;
    if (g_st.n_walks > 0)
      hipLaunchKernelGGL(k_mark_dead_walks, dim3((uint32_t)grid_for(g_st.n_walks)),
                         dim3(kBlock), 0, ctx->stream, g_st.n_walks, g_st.w_start,
                         it->second, g_st.w_dead);
  }
  {
    std::vector<int32_t> starts;
    starts.reserve(n_cv);
    for (int64_t i = 0; i < n_cv; ++i) {
      auto it = g_st.mg2slot.find(created_v[i]);
      if (it != g_st.mg2slot.end()) starts.push_back(it->second);
    }
    MGX_TRY(spawn_walks(ctx, m, g, starts));
  }
  for (int64_t i = 0; i < n_ce; ++i) {
    const int64_t from = created_e[2 * i];
    auto it = g_st.mg2slot.find(from);
    if (it == g_st.mg2slot.end()) continue;
    MGX_TRY(rewire_from(ctx, m, g, it->second, mg2dense.count(from) > 0));
  }
  return compute_rank(ctx, m, out_rank);
}

extern "C" mgx_status mgx_pronline_stats(mgx_context *ctx, int64_t *n_walks,
                                         int64_t *n_live_walks, int64_t *n_live_entries) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  if (n_walks) *n_walks = g_st.n_walks;
  if (n_live_walks && g_st.n_walks > 0) {
    std::vector<uint8_t> dead(g_st.n_walks);
    MGX_HIP_TRY(hipMemcpyAsync(dead.data(), g_st.w_dead, g_st.n_walks,
                               hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    int64_t live = 0;
    for (auto d : dead)
      if (!d) ++live;
    *n_live_walks = live;
  } else if (n_live_walks) {
    *n_live_walks = 0;
  }
  if (n_live_entries) {
    *n_live_entries = 0;
    if (g_st.pool_used > 0) {
      unsigned long long *d = nullptr;
      MGX_TRY(ctx->alloc_async((void **)&d, 8));
      MGX_HIP_TRY(hipMemsetAsync(d, 0, 8, ctx->stream));
      hipLaunchKernelGGL(k_count_live, dim3((uint32_t)grid_for(g_st.pool_used)),
                         dim3(kBlock), 0, ctx->stream, g_st.pool_used, g_st.e_node,
                         g_st.e_walk, g_st.w_dead, d);
      unsigned long long out = 0;
      MGX_HIP_TRY(hipMemcpyAsync(&out, d, 8, hipMemcpyDeviceToHost, ctx->stream));
      MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
      (void)ctx->free_async(d);
      *n_live_entries = (int64_t)out;
    }
  }
  return MGX_OK;
}
