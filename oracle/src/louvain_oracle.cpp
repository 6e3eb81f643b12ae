// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
// Sequential restatement of the grappolo basic Louvain path the reference
// community_detection module runs with coloring=false:
//   - symmetrized CSR build: louvain_alg::GetGrappoloSuitableGraph,
//     /root/reference/src/mage/cpp/community_detection_module/algorithm/louvain.cpp:158-233
//   - phase loop: runMultiPhaseBasic,
//     .../grappolo/BasicCommunitiesDetection/runMultiPhaseBasic.cpp:53-146
//   - Jacobi sweep: parallelLouvianMethod,
//     .../grappolo/BasicCommunitiesDetection/parallelLouvainMethod.cpp:65-290
//   - dQ argmax incl. tie-break and singleton-swap protection: max(),
//     .../grappolo/Utility/utilityClusteringFunctions.cpp:275-310
//   - renumberClustersContiguously: .../grappolo/Utility/buildNextPhase.cpp:49-78
//   - coarsening: buildNextLevelGraphOpt, .../grappolo/Utility/buildNextPhase.cpp:82-
// The restatement is sequential and therefore deterministic; the reference's
// OpenMP build is nondeterministic across thread counts.
// Notable replicated behaviors: result is the assignment BEFORE the last
// sweep (pastCommAss, parallelLouvainMethod.cpp:286); vertices with no edges
// end at community -1 once >= 3 sweeps run in a level (the -1 target of the
// empty-adjacency branch propagates through the past/curr/target rotation).

#include <cassert>
#include <cmath>
#include <cstdint>
#include <map>
#include <numeric>
#include <vector>

#include "../oracle.h"

namespace {

struct SymGraph {
  int64_t nv = 0;
  std::vector<int64_t> ptr;   // [nv+1]
  std::vector<int64_t> col;   // [2E]
  std::vector<double> w;      // [2E]
};

// GetGrappoloSuitableGraph (louvain.cpp:158-233): every input edge stored
// twice (head->tail and tail->head), including self-loops (stored twice in
// the same row).
SymGraph BuildSym(int64_t nv, int64_t ne, const int64_t *src, const int64_t *dst,
                  const double *weights) {
  SymGraph g;
  g.nv = nv;
  g.ptr.assign(nv + 1, 0);
  for (int64_t e = 0; e < ne; ++e) {
    ++g.ptr[src[e] + 1];
    ++g.ptr[dst[e] + 1];
  }
  for (int64_t v = 0; v < nv; ++v) g.ptr[v + 1] += g.ptr[v];
  g.col.resize(2 * ne);
  g.w.resize(2 * ne);
  std::vector<int64_t> cur(g.ptr.begin(), g.ptr.end() - 1);
  for (int64_t e = 0; e < ne; ++e) {
    const double we = weights ? weights[e] : 1.0;
    g.col[cur[src[e]]] = dst[e];
    g.w[cur[src[e]]++] = we;
    g.col[cur[dst[e]]] = src[e];
    g.w[cur[dst[e]]++] = we;
  }
  return g;
}

// parallelLouvianMethod (sequential). Returns the modularity at exit and
// fills C with pastCommAss. `lower` is the Lower parameter (previous phase's
// modularity; parallelLouvainMethod.cpp:252-254 clamps prevMod up to it).
double LouvainLevel(const SymGraph &g, double lower, double thresh, std::vector<int64_t> &C,
                    int64_t *num_iters) {
  const int64_t nv = g.nv;
  // sumVertexDegree (utilityClusteringFunctions.cpp:68-85).
  std::vector<double> vdeg(nv, 0.0);
  for (int64_t v = 0; v < nv; ++v) {
    for (int64_t j = g.ptr[v]; j < g.ptr[v + 1]; ++j) vdeg[v] += g.w[j];
  }
  std::vector<double> cinfo_deg(vdeg);
  std::vector<int64_t> cinfo_size(nv, 1);
  double total = 0.0;
  for (int64_t v = 0; v < nv; ++v) total += vdeg[v];
  const double constant = 1.0 / total;  // calConstantForSecondTerm

  std::vector<int64_t> past(nv), curr(nv), target(nv);
  std::iota(past.begin(), past.end(), 0);
  std::iota(curr.begin(), curr.end(), 0);

  std::vector<double> cupd_deg(nv);
  std::vector<int64_t> cupd_size(nv);
  std::vector<double> cwi(nv);  // clusterWeightInternal

  double prev_mod = -1.0, curr_mod = -1.0;
  int64_t iters = 0;
  while (true) {
    ++iters;
    std::fill(cwi.begin(), cwi.end(), 0.0);
    std::fill(cupd_deg.begin(), cupd_deg.end(), 0.0);
    std::fill(cupd_size.begin(), cupd_size.end(), 0);

    for (int64_t i = 0; i < nv; ++i) {
      const int64_t adj1 = g.ptr[i], adj2 = g.ptr[i + 1];
      if (adj1 == adj2) {
        target[i] = -1;
        continue;
      }
      // buildLocalMapCounter (utilityClusteringFunctions.cpp:171-193):
      // community -> incident weight, own community seeded at 0; self-loop
      // weight recorded separately AND counted in the own bucket.
      std::map<int64_t, double> local;
      double self_loop = 0.0;
      const int64_t sc = curr[i];
      local[sc] = 0.0;
      for (int64_t j = adj1; j < adj2; ++j) {
        if (g.col[j] == i) self_loop += g.w[j];
        local[curr[g.col[j]]] += g.w[j];
      }
      cwi[i] = local[sc];  // Counter[0] == e_i,own

      // max() (utilityClusteringFunctions.cpp:275-310): strict gain argmax,
      // ties (nonzero gain) to the smaller community id, singleton swap
      // protection.
      int64_t max_index = sc;
      double max_gain = 0.0;
      const double eix = local[sc] - self_loop;
      const double ax = cinfo_deg[sc] - vdeg[i];
      for (const auto &[cid, eiy] : local) {
        if (cid == sc) continue;
        const double ay = cinfo_deg[cid];
        const double gain = 2.0 * (eiy - eix) - 2.0 * vdeg[i] * (ay - ax) * constant;
        if (gain > max_gain || (gain == max_gain && gain != 0.0 && cid < max_index)) {
          max_gain = gain;
          max_index = cid;
        }
      }
      if (cinfo_size[max_index] == 1 && cinfo_size[sc] == 1 && max_index > sc) max_index = sc;
      target[i] = max_index;

      if (target[i] != sc && target[i] != -1) {
        cupd_deg[target[i]] += vdeg[i];
        cupd_size[target[i]] += 1;
        cupd_deg[sc] -= vdeg[i];
        cupd_size[sc] -= 1;
      }
    }

    // Modularity from the PRE-update community degrees
    // (parallelLouvainMethod.cpp:238-249).
    double e_xx = 0.0, a2_x = 0.0;
    for (int64_t i = 0; i < nv; ++i) {
      e_xx += cwi[i];
      a2_x += cinfo_deg[i] * cinfo_deg[i];
    }
    curr_mod = e_xx * constant - a2_x * constant * constant;

    if ((curr_mod - prev_mod) < thresh) break;
    prev_mod = curr_mod;
    if (prev_mod < lower) prev_mod = lower;
    for (int64_t i = 0; i < nv; ++i) {
      cinfo_size[i] += cupd_size[i];
      cinfo_deg[i] += cupd_deg[i];
    }
    // Pointer rotation (parallelLouvainMethod.cpp:268-274).
    std::swap(past, curr);
    std::swap(curr, target);
  }

  C = past;
  *num_iters = iters;
  // parallelLouvainMethod.cpp:307 returns prevMod — the (Lower-clamped)
  // modularity of the ADOPTED assignment (pastCommAss) — not the last
  // computed currMod. The difference is < thresh per sweep, but the phase
  // loop compares it against threshold, so returning currMod flips the
  // continue/stop decision at loose thresholds (caught by
  // tests/test_oracle_params.py::test_louvain_threshold_sweep).
  return prev_mod;
}

// renumberClustersContiguously (buildNextPhase.cpp:49-78): first-seen order,
// negative ids kept as-is.
int64_t Renumber(std::vector<int64_t> &C) {
  std::map<int64_t, int64_t> seen;
  int64_t n = 0;
  for (auto &c : C) {
    if (c < 0) continue;
    auto it = seen.find(c);
    if (it != seen.end()) {
      c = it->second;
    } else {
      seen[c] = n;
      c = n++;
    }
  }
  return n;
}

// buildNextLevelGraphOpt (buildNextPhase.cpp:82-): one vertex per cluster; a
// self-loop entry always present in each row (zero weight if no internal
// edges); each inter-cluster pair stored in both rows. Because the input is
// symmetric, internal edges accumulate BOTH directions into the (c,c) entry.
SymGraph Coarsen(const SymGraph &g, const std::vector<int64_t> &C, int64_t num_clusters) {
  std::vector<std::map<int64_t, double>> clu(num_clusters);
  for (int64_t c = 0; c < num_clusters; ++c) clu[c][c] = 0.0;
  for (int64_t i = 0; i < g.nv; ++i) {
    const int64_t ci = C[i];
    for (int64_t j = g.ptr[i]; j < g.ptr[i + 1]; ++j) {
      const int64_t ct = C[g.col[j]];
      if (ci >= ct) clu[ci][ct] += g.w[j];
    }
  }
  SymGraph out;
  out.nv = num_clusters;
  out.ptr.assign(num_clusters + 1, 0);
  for (int64_t c = 0; c < num_clusters; ++c) {
    for (const auto &[d, wv] : clu[c]) {
      ++out.ptr[c + 1];
      if (d != c) ++out.ptr[d + 1];
    }
  }
  for (int64_t c = 0; c < num_clusters; ++c) out.ptr[c + 1] += out.ptr[c];
  out.col.resize(out.ptr[num_clusters]);
  out.w.resize(out.ptr[num_clusters]);
  std::vector<int64_t> cur(out.ptr.begin(), out.ptr.end() - 1);
  for (int64_t c = 0; c < num_clusters; ++c) {
    for (const auto &[d, wv] : clu[c]) {
      out.col[cur[c]] = d;
      out.w[cur[c]++] = wv;
      if (d != c) {
        out.col[cur[d]] = c;
        out.w[cur[d]++] = wv;
      }
    }
  }
  return out;
}

}  // namespace

extern "C" int64_t oracle_louvain(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                  const int64_t *dst, const double *weights, double threshold,
                                  int64_t *out_community) {
  if (n_vertices < 0 || n_edges < 0) return -1;
  for (int64_t v = 0; v < n_vertices; ++v) out_community[v] = -1;
  // The module emits nothing when the scanned graph has no edges
  // (community_detection_module.cpp:72-74).
  if (n_edges == 0 || n_vertices == 0) return 0;
  for (int64_t e = 0; e < n_edges; ++e) {
    if (src[e] < 0 || src[e] >= n_vertices || dst[e] < 0 || dst[e] >= n_vertices) return -1;
  }

  SymGraph g = BuildSym(n_vertices, n_edges, src, dst, weights);

  // runMultiPhaseBasic.cpp:60-146.
  std::vector<int64_t> c_orig(n_vertices, -1);
  double prev_mod = -1.0, curr_mod = -1.0;
  int64_t phase = 1, tot_itr = 0;
  std::vector<int64_t> C;
  while (true) {
    prev_mod = curr_mod;
    int64_t iters = 0;
    curr_mod = LouvainLevel(g, /*lower=*/prev_mod, threshold, C, &iters);
    tot_itr += iters;
    const int64_t num_clusters = Renumber(C);
    if (phase == 1) {
      for (int64_t i = 0; i < n_vertices; ++i) c_orig[i] = C[i];
    } else {
      for (int64_t i = 0; i < n_vertices; ++i) {
        if (c_orig[i] >= 0) c_orig[i] = C[c_orig[i]];
      }
    }
    if (phase > 200 || tot_itr > 100000) break;
    if ((curr_mod - prev_mod) > threshold) {
      g = Coarsen(g, C, num_clusters);
      ++phase;
    } else {
      break;
    }
  }

  int64_t max_c = -1;
  for (int64_t i = 0; i < n_vertices; ++i) {
    out_community[i] = c_orig[i];
    if (c_orig[i] > max_c) max_c = c_orig[i];
  }
  return max_c + 1;
}

extern "C" double oracle_modularity(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                    const int64_t *dst, const double *weights,
                                    const int64_t *community) {
  // Q computed the grappolo way (parallelLouvainMethod.cpp:238-249):
  // Q = e_xx/(2W) - sum_c (deg_c/(2W))^2 over the symmetrized graph.
  if (n_vertices <= 0) return 0.0;
  SymGraph g = BuildSym(n_vertices, n_edges, src, dst, weights);
  std::vector<double> vdeg(n_vertices, 0.0);
  double total = 0.0;
  for (int64_t v = 0; v < n_vertices; ++v) {
    for (int64_t j = g.ptr[v]; j < g.ptr[v + 1]; ++j) vdeg[v] += g.w[j];
    total += vdeg[v];
  }
  if (total == 0.0) return 0.0;
  const double constant = 1.0 / total;
  double e_xx = 0.0;
  std::map<int64_t, double> cdeg;
  for (int64_t v = 0; v < n_vertices; ++v) {
    cdeg[community[v]] += vdeg[v];
    for (int64_t j = g.ptr[v]; j < g.ptr[v + 1]; ++j) {
      if (community[g.col[j]] == community[v]) e_xx += g.w[j];
    }
  }
  double a2 = 0.0;
  for (const auto &[c, d] : cdeg) a2 += d * d;
  return e_xx * constant - a2 * constant * constant;
}
