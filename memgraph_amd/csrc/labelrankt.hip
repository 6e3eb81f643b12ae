// LabelRankT online community detection on gfx950 — replaces the
// reference's LabelRankT (/root/reference/query_modules/
// community_detection_module/algorithm_online/community_detection.cpp):
//   SetStructures (:51-72): per node, label distribution seeded from the
//     self-loop weight plus per-distinct-in-neighbour total edge weights,
//     normalized by sum_w = w_selfloop + sum of in-edge weights.
//   Iteration (:229-269): per candidate node — node selection
//     DistinctEnough (:171-184: the node's most-probable label set is a
//     subset of at most k% of its distinct in-neighbours'), label
//     propagation (:186-203: weighted merge of own (w_selfloop/sum_w) and
//     neighbour (total_w(j,i)/sum_w) distributions), inflation (:205-217:
//     pow(P, exponent) + renormalize), cutoff (:219-227: drop P <
//     min_value); Jacobi application + times_updated bookkeeping.
//   CalculateLabels (:273-303): full or incremental (changed nodes get
//     SetStructures; deleted nodes leave the state) and the
//     none_updated/max_updates stopping rule.
//   AllLabels (:125-150): per-node argmax label (ties to the numerically
//     smallest label id), then labels renumbered 1..k in ascending label-id
//     order; unlabeled nodes get -1.
//
// Deterministic (no RNG); parity bar: identical labels vs the sequential
// oracle restatement (itself pinned exactly against the reference core
// compiled from /root/reference — tests/test_lrt_cpu.py). fp sums are
// reordered vs the reference's unordered_map iteration, so probability
// values agree to ~1e-12 and label argmaxes are compared exactly.
//
// MI355X layout: label distributions live in a device pool of
// (label-slot i32, P f64) pairs with per-node offsets, rebuilt per
// iteration (Jacobi). Node identity is a persistent SLOT keyed by memgraph
// id (host map), so labels — which are node ids in this algorithm —
// survive graph changes. v1 parallelization is one thread per node with a
// per-node scratch segment (hub nodes serialize their merge; the
// LDS-table fast path of louvain.hip is the known next optimization).
// Weights arrive as the graph's f32 edge weights widened to f64 — exact
// for the integral/2^-k weights the tests use (documented).

#include <algorithm>
#include <cstring>
#include <map>
#include <set>
#include <unordered_map>
#include <unordered_set>
#include <vector>

#include <rocprim/rocprim.hpp>

#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

struct LrtState {
  std::unordered_map<int64_t, int32_t> mg2slot;
  std::vector<int64_t> slot2mg;
  std::vector<uint8_t> alive_h;  // node has label state

  // device label distributions: pool of (label slot, P) with per-slot
  // segment [off[s], off[s]+len[s])
  int32_t *lab = nullptr;
  double *p = nullptr;
  uint64_t *off = nullptr;  // [slots_cap]
  uint32_t *len = nullptr;  // [slots_cap]
  int64_t pool_used = 0, pool_cap = 0;
  double *sum_w = nullptr;       // [slots_cap]
  uint32_t *times_upd = nullptr; // [slots_cap]
  int64_t slots_cap = 0;

  // parameters (SetLabels :311-328)
  bool directed = false, weighted = false;
  double sim_th = 0.7, exponent = 4.0, min_value = 0.1, w_selfloop = 1.0;
  int64_t max_iterations = 100, max_updates = 5;
  bool calculated = false;
};

LrtState g_l;

// ---- kernels -------------------------------------------------------------

// Per-node SetStructures (:51-72) over the in-CSR (directed) or sym-CSR
// (undirected): one thread per listed node; cols sorted => distinct
// neighbours are runs. Writes into a fresh pool segment.
struct SetArgs {
  const int32_t *nodes;  // dense ids
  int64_t n;
  const uint32_t *row_ptr;
  const int32_t *col;
  const float *w;  // null => 1.0
  const int32_t *dense2slot;
  double w_selfloop;
  // output segments (counted by k_set_count first)
  const uint64_t *seg_off;  // per listed node
  int32_t *lab;
  double *p;
  uint64_t *slot_off;
  uint32_t *slot_len;
  double *sum_w;
  uint32_t *times_upd;
};

__global__ void k_set_count(int64_t n, const int32_t *nodes, const uint32_t *row_ptr,
                            const int32_t *col, uint32_t *counts) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t v = nodes[i];
    uint32_t distinct = 0;
    int32_t run = -1;
    bool self_seen = false;
    for (uint32_t j = row_ptr[v]; j < row_ptr[v + 1]; ++j) {
      if (col[j] != run) {
        run = col[j];
        ++distinct;
        if (run == v) self_seen = true;
      }
    }
    counts[i] = distinct + (self_seen ? 0u : 1u);  // + own label entry
  }
}

__global__ void k_set_fill(SetArgs A) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < A.n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t v = A.nodes[i];
    const int32_t vs = A.dense2slot[v];
    const uint32_t s = A.row_ptr[v], e = A.row_ptr[v + 1];
    double sw = A.w_selfloop;
    for (uint32_t j = s; j < e; ++j) sw += A.w ? (double)A.w[j] : 1.0;
    const uint64_t base = A.seg_off[i];
    uint64_t out = base;
    // runs over sorted cols; own entry merged with the self-run when present
    int32_t run = -1;
    double run_w = 0.0;
    bool self_emitted = false;
    for (uint32_t j = s; j <= e; ++j) {
      const int32_t c = j < e ? A.col[j] : -2;
      if (c != run) {
        if (run >= 0) {
          double val = run_w / sw;
          if (run == v) {
            val += A.w_selfloop / sw;  // node_label_Ps[v] = selfloop + self-edges
            self_emitted = true;
          }
          A.lab[out] = A.dense2slot[run];
          A.p[out] = val;
          ++out;
        }
        run = c;
        run_w = 0.0;
      }
      if (j < e) run_w += A.w ? (double)A.w[j] : 1.0;
    }
    if (!self_emitted) {
      A.lab[out] = vs;
      A.p[out] = A.w_selfloop / sw;
      ++out;
    }
    A.slot_off[vs] = base;
    A.slot_len[vs] = (uint32_t)(out - base);
    A.sum_w[vs] = sw;
    A.times_upd[vs] = 0;
  }
}

// Per-node candidate-size bound for the propagate merge: len(i) + per-EDGE
// sum of len(src). (A bound, not the exact distinct-label count.)
__global__ void k_merge_bound(int64_t n, const int32_t *nodes, const uint32_t *row_ptr,
                              const int32_t *col, const int32_t *dense2slot,
                              const uint32_t *slot_len, uint32_t *bound) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t v = nodes[i];
    uint64_t b = slot_len[dense2slot[v]];
    for (uint32_t j = row_ptr[v]; j < row_ptr[v + 1]; ++j)
      b += slot_len[dense2slot[col[j]]];
    bound[i] = (uint32_t)(b < 0xFFFFFFFFull ? b : 0xFFFFFFFFull);
  }
}

// DistinctEnough (:171-184): one thread per candidate node.
__global__ void k_distinct(int64_t n, const int32_t *nodes, const uint32_t *row_ptr,
                           const int32_t *col, const int32_t *dense2slot,
                           const int32_t *lab, const double *p, const uint64_t *off,
                           const uint32_t *len, double sim_th, uint8_t *selected) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t v = nodes[i];
    const int32_t vs = dense2slot[v];
    // max P of v
    double maxp = 0.0;
    for (uint32_t k = 0; k < len[vs]; ++k) {
      const double pv = p[off[vs] + k];
      if (pv > maxp) maxp = pv;
    }
    uint32_t similar = 0, distinct_nb = 0;
    int32_t run = -1;
    for (uint32_t j = row_ptr[v]; j < row_ptr[v + 1]; ++j) {
      const int32_t c = col[j];
      if (c == run) continue;
      run = c;
      ++distinct_nb;
      const int32_t js = dense2slot[c];
      // max P of j
      double maxpj = 0.0;
      for (uint32_t k = 0; k < len[js]; ++k) {
        const double pj = p[off[js] + k];
        if (pj > maxpj) maxpj = pj;
      }
      // MP(v) subset of MP(j)?
      bool subset = true;
      for (uint32_t k = 0; k < len[vs] && subset; ++k) {
        if (p[off[vs] + k] != maxp) continue;
        const int32_t l = lab[off[vs] + k];
        bool found = false;
        for (uint32_t m = 0; m < len[js]; ++m) {
          if (lab[off[js] + m] == l && p[off[js] + m] == maxpj) {
            found = true;
            break;
          }
        }
        if (!found) subset = false;
      }
      if (subset) ++similar;
    }
    selected[i] = (double)similar <= (double)distinct_nb * sim_th ? 1 : 0;
  }
}

// Propagate + Inflate + Cutoff (:186-227) for selected nodes into scratch
// segments; emits the surviving count.
struct PropArgs {
  const int32_t *nodes;
  int64_t n;
  const uint8_t *selected;
  const uint32_t *row_ptr;
  const int32_t *col;
  const float *w;
  const int32_t *dense2slot;
  const int32_t *lab;
  const double *p;
  const uint64_t *off;
  const uint32_t *len;
  const double *sum_w;
  double w_selfloop, exponent, min_value;
  // scratch segments per listed node
  const uint64_t *scr_off;
  int32_t *scr_lab;
  double *scr_p;
  uint32_t *out_count;  // surviving labels per listed node (0 if unselected)
};

__global__ void k_propagate(PropArgs A) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < A.n;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (!A.selected[i]) {
      A.out_count[i] = 0;
      continue;
    }
    const int32_t v = A.nodes[i];
    const int32_t vs = A.dense2slot[v];
    int32_t *tl = A.scr_lab + A.scr_off[i];
    double *tp = A.scr_p + A.scr_off[i];
    uint32_t used = 0;
    auto add = [&](int32_t l, double val) {
      for (uint32_t k = 0; k < used; ++k) {
        if (tl[k] == l) {
          tp[k] += val;
          return;
        }
      }
      tl[used] = l;
      tp[used] = val;
      ++used;
    };
    // own labels (:190-192)
    const double own_f = A.w_selfloop / A.sum_w[vs];
    for (uint32_t k = 0; k < A.len[vs]; ++k)
      add(A.lab[A.off[vs] + k], own_f * A.p[A.off[vs] + k]);
    // per-edge neighbour contributions (== per-distinct-neighbour
    // total-weight contributions, :195-199)
    for (uint32_t j = A.row_ptr[v]; j < A.row_ptr[v + 1]; ++j) {
      const int32_t js = A.dense2slot[A.col[j]];
      const double f = (A.w ? (double)A.w[j] : 1.0) / A.sum_w[vs];
      for (uint32_t k = 0; k < A.len[js]; ++k)
        add(A.lab[A.off[js] + k], f * A.p[A.off[js] + k]);
    }
    // Inflate (:205-217)
    double sum_ps = 0.0;
    for (uint32_t k = 0; k < used; ++k) {
      tp[k] = pow(tp[k], A.exponent);
      sum_ps += tp[k];
    }
    // Cutoff (:219-227) with compaction
    uint32_t kept = 0;
    for (uint32_t k = 0; k < used; ++k) {
      const double val = tp[k] / sum_ps;
      if (val >= A.min_value) {
        tl[kept] = tl[k];
        tp[kept] = val;
        ++kept;
      }
    }
    A.out_count[i] = kept;
  }
}

// Build the next pool: unlisted/unselected slots copy their old segment;
// selected listed slots take the scratch segment. One thread per slot.
struct RebuildArgs {
  int64_t n_slots;
  const int32_t *slot_sel_idx;  // [n_slots]: listed-node index if selected, else -1
  const int32_t *old_lab;
  const double *old_p;
  const uint64_t *old_off;
  const uint32_t *old_len;
  const uint64_t *scr_off;
  const int32_t *scr_lab;
  const double *scr_p;
  const uint32_t *scr_cnt;
  const uint64_t *new_off;  // per slot (exclusive scan of new lens)
  int32_t *new_lab;
  double *new_p;
  uint64_t *out_off;
  uint32_t *out_len;
  uint32_t *times_upd;
  unsigned long long *most_upd;  // max times_updated among applied
  uint32_t *any_updated;
};

__global__ void k_rebuild(RebuildArgs A) {
  for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < A.n_slots;
       s += (int64_t)gridDim.x * blockDim.x) {
    const int32_t li = A.slot_sel_idx[s];
    const uint64_t dst = A.new_off[s];
    if (li >= 0) {
      const uint32_t n = A.scr_cnt[li];
      for (uint32_t k = 0; k < n; ++k) {
        A.new_lab[dst + k] = A.scr_lab[A.scr_off[li] + k];
        A.new_p[dst + k] = A.scr_p[A.scr_off[li] + k];
      }
      A.out_off[s] = dst;
      A.out_len[s] = n;
      const uint32_t t = ++A.times_upd[s];
      atomicMax(A.most_upd, (unsigned long long)t);
      atomicOr(A.any_updated, 1u);
    } else {
      const uint32_t n = A.old_len[s];
      for (uint32_t k = 0; k < n; ++k) {
        A.new_lab[dst + k] = A.old_lab[A.old_off[s] + k];
        A.new_p[dst + k] = A.old_p[A.old_off[s] + k];
      }
      A.out_off[s] = dst;
      A.out_len[s] = n;
    }
  }
}

// AllLabels argmax (:111-123): label with max P, ties to the smaller
// MEMGRAPH id of the label (labels are node ids).
__global__ void k_argmax(int64_t n_slots, const int32_t *lab, const double *p,
                         const uint64_t *off, const uint32_t *len,
                         const int64_t *slot2mg, int64_t *out_label_mg) {
  for (int64_t s = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; s < n_slots;
       s += (int64_t)gridDim.x * blockDim.x) {
    double maxp = 0.0;
    int64_t best = -1;
    for (uint32_t k = 0; k < len[s]; ++k) {
      const double pv = p[off[s] + k];
      const int64_t lmg = slot2mg[lab[off[s] + k]];
      if (pv > maxp || (pv == maxp && best >= 0 && lmg < best)) {
        maxp = pv;
        best = lmg;
      }
    }
    out_label_mg[s] = best;
  }
}

}  // namespace

// ---- host orchestration --------------------------------------------------

namespace {

void lrt_free() {
  if (g_l.lab) (void)hipFree(g_l.lab);
  if (g_l.p) (void)hipFree(g_l.p);
  if (g_l.off) (void)hipFree(g_l.off);
  if (g_l.len) (void)hipFree(g_l.len);
  if (g_l.sum_w) (void)hipFree(g_l.sum_w);
  if (g_l.times_upd) (void)hipFree(g_l.times_upd);
  g_l = LrtState{};
}

int32_t lrt_slot(int64_t mg) {
  auto it = g_l.mg2slot.find(mg);
  if (it != g_l.mg2slot.end()) return it->second;
  const int32_t s = (int32_t)g_l.slot2mg.size();
  g_l.mg2slot.emplace(mg, s);
  g_l.slot2mg.push_back(mg);
  g_l.alive_h.push_back(0);
  return s;
}

mgx_status lrt_grow_slots(mgx_context *ctx, int64_t need) {
  if (need <= g_l.slots_cap) return MGX_OK;
  int64_t cap = g_l.slots_cap > 0 ? g_l.slots_cap : 256;
  while (cap < need) cap *= 2;
  uint64_t *noff = nullptr;
  uint32_t *nlen = nullptr, *ntu = nullptr;
  double *nsw = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&noff, cap * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&nlen, cap * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&ntu, cap * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&nsw, cap * 8));
  MGX_HIP_TRY(hipMemsetAsync(nlen, 0, cap * 4, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(ntu, 0, cap * 4, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(noff, 0, cap * 8, ctx->stream));
  if (g_l.slots_cap > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(noff, g_l.off, g_l.slots_cap * 8,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(nlen, g_l.len, g_l.slots_cap * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(ntu, g_l.times_upd, g_l.slots_cap * 4,
                               hipMemcpyDeviceToDevice, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(nsw, g_l.sum_w, g_l.slots_cap * 8,
                               hipMemcpyDeviceToDevice, ctx->stream));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  if (g_l.off) (void)hipFree(g_l.off);
  if (g_l.len) (void)hipFree(g_l.len);
  if (g_l.times_upd) (void)hipFree(g_l.times_upd);
  if (g_l.sum_w) (void)hipFree(g_l.sum_w);
  g_l.off = noff;
  g_l.len = nlen;
  g_l.times_upd = ntu;
  g_l.sum_w = nsw;
  g_l.slots_cap = cap;
  return MGX_OK;
}

// Exclusive scan of u32 into u64 offsets (host round-trip is fine at these
// list sizes; the pool itself stays on device).
mgx_status scan_counts_u64(mgx_context *ctx, const uint32_t *d_counts, int64_t n,
                           std::vector<uint64_t> *off_h, uint64_t *total) {
  std::vector<uint32_t> c(n > 0 ? n : 1);
  if (n > 0) {
    MGX_HIP_TRY(hipMemcpyAsync(c.data(), d_counts, n * 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }
  off_h->assign(n > 0 ? n : 1, 0);
  uint64_t acc = 0;
  for (int64_t i = 0; i < n; ++i) {
    (*off_h)[i] = acc;
    acc += c[i];
  }
  *total = acc;
  return MGX_OK;
}

struct LrtMaps {
  int32_t *dense2slot = nullptr;
  std::vector<int32_t> d2s_h;
  int64_t V = 0;
  ~LrtMaps() {
    if (dense2slot) (void)hipFree(dense2slot);
  }
};

mgx_status lrt_build_maps(mgx_context *ctx, int64_t V, const int64_t *dense_to_mg,
                          LrtMaps *m) {
  m->V = V;
  m->d2s_h.assign(V > 0 ? V : 1, -1);
  for (int64_t v = 0; v < V; ++v) m->d2s_h[v] = lrt_slot(dense_to_mg[v]);
  MGX_TRY(lrt_grow_slots(ctx, (int64_t)g_l.slot2mg.size()));
  MGX_HIP_TRY(mgx_hip_malloc(&m->dense2slot, (V > 0 ? V : 1) * 4));
  MGX_HIP_TRY(hipMemcpyAsync(m->dense2slot, m->d2s_h.data(), (V > 0 ? V : 1) * 4,
                             hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  return MGX_OK;
}

// SetStructures for the listed dense nodes, appending fresh segments to a
// NEW pool region (old segments of other nodes stay valid: the pool only
// grows here; a full rebuild happens per iteration anyway).
mgx_status lrt_set_structures(mgx_context *ctx, mgx_graph *g, const LrtMaps &m,
                              const std::vector<int32_t> &nodes_dense,
                              const uint32_t *row_ptr, const int32_t *col,
                              const float *w) {
  const int64_t n = (int64_t)nodes_dense.size();
  if (n == 0) return MGX_OK;
  int32_t *d_nodes = nullptr;
  uint32_t *d_counts = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_nodes, n * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&d_counts, n * 4));
  MGX_HIP_TRY(hipMemcpyAsync(d_nodes, nodes_dense.data(), n * 4, hipMemcpyHostToDevice,
                             ctx->stream));
  hipLaunchKernelGGL(k_set_count, dim3((uint32_t)grid_for(n)), dim3(kBlock), 0,
                     ctx->stream, n, d_nodes, row_ptr, col, d_counts);
  std::vector<uint64_t> seg_off;
  uint64_t total = 0;
  MGX_TRY(scan_counts_u64(ctx, d_counts, n, &seg_off, &total));
  // append region
  const int64_t need = g_l.pool_used + (int64_t)total;
  if (need > g_l.pool_cap) {
    int64_t cap = g_l.pool_cap > 0 ? g_l.pool_cap : 4096;
    while (cap < need) cap *= 2;
    int32_t *nl = nullptr;
    double *np = nullptr;
    MGX_HIP_TRY(mgx_hip_malloc(&nl, cap * 4));
    MGX_HIP_TRY(mgx_hip_malloc(&np, cap * 8));
    if (g_l.pool_used > 0) {
      MGX_HIP_TRY(hipMemcpyAsync(nl, g_l.lab, g_l.pool_used * 4,
                                 hipMemcpyDeviceToDevice, ctx->stream));
      MGX_HIP_TRY(hipMemcpyAsync(np, g_l.p, g_l.pool_used * 8, hipMemcpyDeviceToDevice,
                                 ctx->stream));
    }
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (g_l.lab) (void)hipFree(g_l.lab);
    if (g_l.p) (void)hipFree(g_l.p);
    g_l.lab = nl;
    g_l.p = np;
    g_l.pool_cap = cap;
  }
  for (auto &o : seg_off) o += (uint64_t)g_l.pool_used;
  uint64_t *d_seg = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_seg, n * 8));
  MGX_HIP_TRY(hipMemcpyAsync(d_seg, seg_off.data(), n * 8, hipMemcpyHostToDevice,
                             ctx->stream));
  SetArgs A;
  A.nodes = d_nodes;
  A.n = n;
  A.row_ptr = row_ptr;
  A.col = col;
  A.w = w;
  A.dense2slot = m.dense2slot;
  A.w_selfloop = g_l.w_selfloop;
  A.seg_off = d_seg;
  A.lab = g_l.lab;
  A.p = g_l.p;
  A.slot_off = g_l.off;
  A.slot_len = g_l.len;
  A.sum_w = g_l.sum_w;
  A.times_upd = g_l.times_upd;
  hipLaunchKernelGGL(k_set_fill, dim3((uint32_t)grid_for(n)), dim3(kBlock), 0, ctx->stream,
                     A);
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  g_l.pool_used += (int64_t)total;
  (void)hipFree(d_nodes);
  (void)hipFree(d_counts);
  (void)hipFree(d_seg);
  return MGX_OK;
}

// One Iteration (:229-269). candidates = dense nodes passing the
// incremental filter. Returns (none_updated, most_updates).
mgx_status lrt_iteration(mgx_context *ctx, const LrtMaps &m, const uint32_t *row_ptr,
                         const int32_t *col, const float *w,
                         const std::vector<int32_t> &candidates, bool *none_updated,
                         uint64_t *most_updates) {
  const int64_t n = (int64_t)candidates.size();
  const int64_t n_slots = (int64_t)g_l.slot2mg.size();
  *none_updated = true;
  *most_updates = 0;
  if (n == 0) return MGX_OK;
  int32_t *d_nodes = nullptr;
  uint8_t *d_sel = nullptr;
  uint32_t *d_bound = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_nodes, n * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&d_sel, n));
  MGX_HIP_TRY(mgx_hip_malloc(&d_bound, n * 4));
  MGX_HIP_TRY(hipMemcpyAsync(d_nodes, candidates.data(), n * 4, hipMemcpyHostToDevice,
                             ctx->stream));
  hipLaunchKernelGGL(k_distinct, dim3((uint32_t)grid_for(n)), dim3(kBlock), 0, ctx->stream,
                     n, d_nodes, row_ptr, col, m.dense2slot, g_l.lab, g_l.p, g_l.off,
                     g_l.len, g_l.sim_th, d_sel);
  hipLaunchKernelGGL(k_merge_bound, dim3((uint32_t)grid_for(n)), dim3(kBlock), 0,
                     ctx->stream, n, d_nodes, row_ptr, col, m.dense2slot, g_l.len,
                     d_bound);
  // zero bounds for unselected nodes to keep scratch small
  {
    std::vector<uint8_t> sel(n);
    std::vector<uint32_t> bound(n);
    MGX_HIP_TRY(hipMemcpyAsync(sel.data(), d_sel, n, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(bound.data(), d_bound, n * 4, hipMemcpyDeviceToHost,
                               ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    for (int64_t i = 0; i < n; ++i)
      if (!sel[i]) bound[i] = 0;
    MGX_HIP_TRY(hipMemcpyAsync(d_bound, bound.data(), n * 4, hipMemcpyHostToDevice,
                               ctx->stream));
  }
  std::vector<uint64_t> scr_off;
  uint64_t scr_total = 0;
  MGX_TRY(scan_counts_u64(ctx, d_bound, n, &scr_off, &scr_total));
  int32_t *scr_lab = nullptr;
  double *scr_p = nullptr;
  uint64_t *d_scr_off = nullptr;
  uint32_t *d_cnt = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&scr_lab, (scr_total > 0 ? scr_total : 1) * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&scr_p, (scr_total > 0 ? scr_total : 1) * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_scr_off, n * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_cnt, n * 4));
  MGX_HIP_TRY(hipMemcpyAsync(d_scr_off, scr_off.data(), n * 8, hipMemcpyHostToDevice,
                             ctx->stream));
  PropArgs P;
  P.nodes = d_nodes;
  P.n = n;
  P.selected = d_sel;
  P.row_ptr = row_ptr;
  P.col = col;
  P.w = w;
  P.dense2slot = m.dense2slot;
  P.lab = g_l.lab;
  P.p = g_l.p;
  P.off = g_l.off;
  P.len = g_l.len;
  P.sum_w = g_l.sum_w;
  P.w_selfloop = g_l.w_selfloop;
  P.exponent = g_l.exponent;
  P.min_value = g_l.min_value;
  P.scr_off = d_scr_off;
  P.scr_lab = scr_lab;
  P.scr_p = scr_p;
  P.out_count = d_cnt;
  hipLaunchKernelGGL(k_propagate, dim3((uint32_t)grid_for(n)), dim3(kBlock), 0,
                     ctx->stream, P);

  // slot -> listed-node index (selected only)
  std::vector<int32_t> slot_sel(n_slots, -1);
  std::vector<uint8_t> sel(n);
  std::vector<uint32_t> cnt(n);
  MGX_HIP_TRY(hipMemcpyAsync(sel.data(), d_sel, n, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipMemcpyAsync(cnt.data(), d_cnt, n * 4, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  bool any = false;
  for (int64_t i = 0; i < n; ++i) {
    if (sel[i]) {
      slot_sel[m.d2s_h[candidates[i]]] = (int32_t)i;
      any = true;
    }
  }
  *none_updated = !any;
  if (!any) {
    (void)hipFree(d_nodes);
    (void)hipFree(d_sel);
    (void)hipFree(d_bound);
    (void)hipFree(scr_lab);
    (void)hipFree(scr_p);
    (void)hipFree(d_scr_off);
    (void)hipFree(d_cnt);
    return MGX_OK;
  }

  // new pool = sum over slots of new lens
  std::vector<uint32_t> old_len(n_slots);
  MGX_HIP_TRY(hipMemcpyAsync(old_len.data(), g_l.len, n_slots * 4, hipMemcpyDeviceToHost,
                             ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  std::vector<uint64_t> new_off(n_slots);
  uint64_t total = 0;
  for (int64_t s = 0; s < n_slots; ++s) {
    new_off[s] = total;
    total += slot_sel[s] >= 0 ? cnt[slot_sel[s]] : old_len[s];
  }
  int32_t *nlab = nullptr;
  double *np = nullptr;
  uint64_t *d_new_off = nullptr;
  int32_t *d_slot_sel = nullptr;
  uint64_t *d_out_off = nullptr;
  uint32_t *d_out_len = nullptr;
  unsigned long long *d_most = nullptr;
  uint32_t *d_any = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&nlab, (total > 0 ? total : 1) * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&np, (total > 0 ? total : 1) * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_new_off, n_slots * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_slot_sel, n_slots * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&d_out_off, n_slots * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_out_len, n_slots * 4));
  MGX_HIP_TRY(mgx_hip_malloc(&d_most, 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_any, 4));
  MGX_HIP_TRY(hipMemcpyAsync(d_new_off, new_off.data(), n_slots * 8,
                             hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipMemcpyAsync(d_slot_sel, slot_sel.data(), n_slots * 4,
                             hipMemcpyHostToDevice, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(d_most, 0, 8, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(d_any, 0, 4, ctx->stream));
  RebuildArgs R;
  R.n_slots = n_slots;
  R.slot_sel_idx = d_slot_sel;
  R.old_lab = g_l.lab;
  R.old_p = g_l.p;
  R.old_off = g_l.off;
  R.old_len = g_l.len;
  R.scr_off = d_scr_off;
  R.scr_lab = scr_lab;
  R.scr_p = scr_p;
  R.scr_cnt = d_cnt;
  R.new_off = d_new_off;
  R.new_lab = nlab;
  R.new_p = np;
  R.out_off = d_out_off;
  R.out_len = d_out_len;
  R.times_upd = g_l.times_upd;
  R.most_upd = d_most;
  R.any_updated = d_any;
  hipLaunchKernelGGL(k_rebuild, dim3((uint32_t)grid_for(n_slots)), dim3(kBlock), 0,
                     ctx->stream, R);
  unsigned long long most = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&most, d_most, 8, hipMemcpyDeviceToHost, ctx->stream));
  // swap pool + off/len
  MGX_HIP_TRY(hipMemcpyAsync(g_l.off, d_out_off, n_slots * 8, hipMemcpyDeviceToDevice,
                             ctx->stream));
  MGX_HIP_TRY(hipMemcpyAsync(g_l.len, d_out_len, n_slots * 4, hipMemcpyDeviceToDevice,
                             ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  (void)hipFree(g_l.lab);
  (void)hipFree(g_l.p);
  g_l.lab = nlab;
  g_l.p = np;
  g_l.pool_used = (int64_t)total;
  g_l.pool_cap = (int64_t)(total > 0 ? total : 1);
  *most_updates = most;

  (void)hipFree(d_nodes);
  (void)hipFree(d_sel);
  (void)hipFree(d_bound);
  (void)hipFree(scr_lab);
  (void)hipFree(scr_p);
  (void)hipFree(d_scr_off);
  (void)hipFree(d_cnt);
  (void)hipFree(d_new_off);
  (void)hipFree(d_slot_sel);
  (void)hipFree(d_out_off);
  (void)hipFree(d_out_len);
  (void)hipFree(d_most);
  (void)hipFree(d_any);
  return MGX_OK;
}

// AllLabels (:125-150) into out_label (by dense id), renumbered 1..k.
mgx_status lrt_all_labels(mgx_context *ctx, const LrtMaps &m, int64_t *out_label) {
  const int64_t n_slots = (int64_t)g_l.slot2mg.size();
  if (n_slots == 0 || m.V == 0) return MGX_OK;
  int64_t *d_s2mg = nullptr, *d_raw = nullptr;
  MGX_HIP_TRY(mgx_hip_malloc(&d_s2mg, n_slots * 8));
  MGX_HIP_TRY(mgx_hip_malloc(&d_raw, n_slots * 8));
  MGX_HIP_TRY(hipMemcpyAsync(d_s2mg, g_l.slot2mg.data(), n_slots * 8,
                             hipMemcpyHostToDevice, ctx->stream));
  hipLaunchKernelGGL(k_argmax, dim3((uint32_t)grid_for(n_slots)), dim3(kBlock), 0,
                     ctx->stream, n_slots, g_l.lab, g_l.p, g_l.off, g_l.len, d_s2mg,
                     d_raw);
  std::vector<int64_t> raw(n_slots);
  MGX_HIP_TRY(hipMemcpyAsync(raw.data(), d_raw, n_slots * 8, hipMemcpyDeviceToHost,
                             ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  (void)hipFree(d_s2mg);
  (void)hipFree(d_raw);
  // renumber in ascending label-mg-id order (:135-141); only ALIVE slots
  std::set<int64_t> ordered;
  for (int64_t s = 0; s < n_slots; ++s)
    if (g_l.alive_h[s] && raw[s] >= 0) ordered.insert(raw[s]);
  std::unordered_map<int64_t, int64_t> lookup;
  int64_t li = 1;
  for (auto l : ordered) lookup[l] = li++;
  for (int64_t v = 0; v < m.V; ++v) {
    const int32_t s = m.d2s_h[v];
    out_label[v] = (g_l.alive_h[s] && raw[s] >= 0) ? lookup[raw[s]] : -1;
  }
  return MGX_OK;
}

// CalculateLabels (:273-303)
mgx_status lrt_calculate(mgx_context *ctx, mgx_graph *g, const LrtMaps &m,
                         const std::set<int64_t> &changed_mg,
                         const std::set<int64_t> &to_delete_mg, bool persist,
                         int64_t *out_label) {
  // directed: identity-layout in-CSR (MGX_BUILD_NO_PERM; weights in in_w);
  // undirected: sym-CSR (weights in sym_w)
  const uint32_t *row_ptr = g_l.directed ? g->in_row_ptr : g->sym_row_ptr;
  const int32_t *col = g_l.directed ? g->in_col : g->sym_col;
  const float *w = g_l.weighted ? (g_l.directed ? g->in_w : g->sym_w) : nullptr;

  const bool incremental = !changed_mg.empty();
  std::vector<int32_t> nodes_all(m.V);
  for (int64_t v = 0; v < m.V; ++v) nodes_all[v] = (int32_t)v;

  if (incremental) {
    for (auto mg : to_delete_mg) {  // RemoveDeletedNodes (:74-80)
      auto it = g_l.mg2slot.find(mg);
      if (it != g_l.mg2slot.end()) g_l.alive_h[it->second] = 0;
    }
    std::vector<int32_t> changed_dense;
    std::unordered_map<int64_t, int32_t> mg2dense;
    for (int64_t v = 0; v < m.V; ++v) mg2dense.emplace(g_l.slot2mg[m.d2s_h[v]], (int32_t)v);
    for (auto mg : changed_mg) {
      auto it = mg2dense.find(mg);
      if (it != mg2dense.end()) {
        changed_dense.push_back(it->second);
        g_l.alive_h[m.d2s_h[it->second]] = 1;
      }
    }
    MGX_TRY(lrt_set_structures(ctx, g, m, changed_dense, row_ptr, col, w));
    // candidates: changed and not deleted (:238-243)
    std::vector<int32_t> cands = changed_dense;
    for (int64_t it = 0; it < g_l.max_iterations; ++it) {
      bool none = true;
      uint64_t most = 0;
      MGX_TRY(lrt_iteration(ctx, m, row_ptr, col, w, cands, &none, &most));
      if (none || most > (uint64_t)g_l.max_updates) break;
    }
  } else {
    // full recompute (:285-292)
    g_l.pool_used = 0;
    for (auto &a : g_l.alive_h) a = 0;
    for (int64_t v = 0; v < m.V; ++v) g_l.alive_h[m.d2s_h[v]] = 1;
    MGX_HIP_TRY(hipMemsetAsync(g_l.len, 0, g_l.slots_cap * 4, ctx->stream));
    MGX_HIP_TRY(hipMemsetAsync(g_l.times_upd, 0, g_l.slots_cap * 4, ctx->stream));
    MGX_TRY(lrt_set_structures(ctx, g, m, nodes_all, row_ptr, col, w));
    for (int64_t it = 0; it < g_l.max_iterations; ++it) {
      bool none = true;
      uint64_t most = 0;
      MGX_TRY(lrt_iteration(ctx, m, row_ptr, col, w, nodes_all, &none, &most));
      if (none || most > (uint64_t)g_l.max_updates) break;
    }
  }
  if (persist) g_l.calculated = true;
  // ResetTimesUpdated (:300)
  MGX_HIP_TRY(hipMemsetAsync(g_l.times_upd, 0, g_l.slots_cap * 4, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  return lrt_all_labels(ctx, m, out_label);
}

}  // namespace

extern "C" int mgx_lrt_initialized(void) { return g_l.calculated ? 1 : 0; }

extern "C" mgx_status mgx_lrt_reset(mgx_context *ctx) {
  (void)ctx;
  lrt_free();
  return MGX_OK;
}

static mgx_status lrt_check_graph(mgx_graph *g, bool directed) {
  if (!g) return MGX_OK;
  if (directed) {
    if (!((g->flags & MGX_BUILD_IN_CSR) && (g->flags & MGX_BUILD_NO_PERM))) {
      mgx_set_error("lrt directed needs IN_CSR|NO_PERM");
      return MGX_ERR_INVALID_ARGUMENT;
    }
  } else if (!(g->flags & MGX_BUILD_SYM_CSR)) {
    mgx_set_error("lrt undirected needs SYM_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  return MGX_OK;
}

extern "C" mgx_status mgx_lrt_set(mgx_context *ctx, mgx_graph *g,
                                  const int64_t *dense_to_mg, int directed, int weighted,
                                  double similarity_threshold, double exponent,
                                  double min_value, double w_selfloop,
                                  int64_t max_iterations, int64_t max_updates,
                                  int64_t *out_label) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  MGX_TRY(lrt_check_graph(g, directed != 0));
  lrt_free();
  g_l.directed = directed != 0;
  g_l.weighted = weighted != 0;
  g_l.sim_th = similarity_threshold;
  g_l.exponent = exponent;
  g_l.min_value = min_value;
  g_l.w_selfloop = w_selfloop;
  g_l.max_iterations = max_iterations;
  g_l.max_updates = max_updates;
  const int64_t V = g ? g->n_vertices : 0;
  LrtMaps m;
  MGX_TRY(lrt_build_maps(ctx, V, dense_to_mg, &m));
  if (V == 0) {
    g_l.calculated = true;
    return MGX_OK;
  }
  return lrt_calculate(ctx, g, m, {}, {}, /*persist=*/true, out_label);
}

extern "C" mgx_status mgx_lrt_get(mgx_context *ctx, mgx_graph *g,
                                  const int64_t *dense_to_mg, int64_t *out_label,
                                  int *ran_set) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  const int64_t V = g ? g->n_vertices : 0;
  LrtMaps m;
  MGX_TRY(lrt_build_maps(ctx, V, dense_to_mg, &m));
  if (!g_l.calculated) {
    // GetLabels on an uncalculated state: full compute, NOT persisted
    // (:305-309 — persist=false)
    if (ran_set) *ran_set = 1;
    if (V == 0) return MGX_OK;
    return lrt_calculate(ctx, g, m, {}, {}, /*persist=*/false, out_label);
  }
  if (ran_set) *ran_set = 0;
  if (V == 0) return MGX_OK;
  return lrt_all_labels(ctx, m, out_label);
}

extern "C" mgx_status mgx_lrt_update(mgx_context *ctx, mgx_graph *g,
                                     const int64_t *dense_to_mg, const int64_t *mod_v,
                                     int64_t n_mv, const int64_t *mod_e, int64_t n_me,
                                     const int64_t *del_v, int64_t n_dv,
                                     const int64_t *del_e, int64_t n_de,
                                     int64_t *out_label) {
  MGX_HIP_TRY(hipSetDevice(ctx->device));
  const int64_t V = g ? g->n_vertices : 0;
  LrtMaps m;
  MGX_TRY(lrt_build_maps(ctx, V, dense_to_mg, &m));
  if (!g_l.calculated) {
    if (V == 0) return MGX_OK;
    return lrt_calculate(ctx, g, m, {}, {}, /*persist=*/false, out_label);
  }
  // UpdateLabels (:330-351)
  std::set<int64_t> changed(mod_v, mod_v + n_mv);
  std::set<int64_t> to_delete(del_v, del_v + n_dv);
  for (int64_t i = 0; i < n_me; ++i) {
    changed.insert(mod_e[2 * i]);
    changed.insert(mod_e[2 * i + 1]);
  }
  for (int64_t i = 0; i < n_de; ++i) {
    if (!to_delete.count(del_e[2 * i])) changed.insert(del_e[2 * i]);
    if (!to_delete.count(del_e[2 * i + 1])) changed.insert(del_e[2 * i + 1]);
  }
  return lrt_calculate(ctx, g, m, changed, to_delete, /*persist=*/true, out_label);
}
