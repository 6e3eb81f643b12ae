// ORACLE/_REF — TEST INFRASTRUCTURE ONLY (see ref_wrap.cpp header).
// Drives the reference's ONLINE algorithm cores, compiled from the sources
// where they lie under /root/reference/query_modules (never copied):
//   - pagerank_online_alg (pagerank_module/algorithm_online/pagerank.cpp)
//     — random_device-seeded inside the reference, so tests can pin only
//     DISTRIBUTIONS against it (DESIGN.md statistical-parity bar level 2);
//   - katz_alg online (katz_centrality_module/algorithm/katz.cpp) —
//     deterministic, exact pin;
//   - LabelRankT (community_detection_module/algorithm_online/
//     community_detection.cpp) — deterministic, exact pin.
// Built into a SEPARATE libref_online.so (the online katz shares katz_alg
// symbol names with the MAGE copy in libref.so).
//
// Graphs arrive as dense node-id lists + (src,dst) edge lists and are
// loaded into mg_graph::Graph exactly as mg_utility::GetGraphView would
// (include/mg_utils.hpp:127-150).

#include <cstdint>
#include <memory>
#include <vector>

#include <mg_graph.hpp>

#include "algorithm_online/pagerank.hpp"
#include "katz.hpp"  // the ONLINE katz (query_modules copy)
#include "algorithm_online/community_detection.hpp"

namespace {

std::unique_ptr<mg_graph::Graph<>> build_graph(int64_t n_nodes, const int64_t *nodes,
                                               int64_t n_edges, const int64_t *src,
                                               const int64_t *dst,
                                               mg_graph::GraphType type,
                                               const double *weights = nullptr) {
  auto g = std::make_unique<mg_graph::Graph<>>();
  for (int64_t v = 0; v < n_nodes; ++v) g->CreateNode((uint64_t)nodes[v]);
  for (int64_t e = 0; e < n_edges; ++e) {
    // inner edge id defaults to creation index (mg_graph.hpp:198-204)
    g->CreateEdge((uint64_t)src[e], (uint64_t)dst[e], type, std::nullopt,
                  weights != nullptr, weights ? weights[e] : 0.0);
  }
  return g;
}

void fill_rank(const std::vector<std::pair<uint64_t, double>> &res, int64_t n_nodes,
               const int64_t *nodes, double *out) {
  // out is indexed by position in `nodes`
  for (int64_t i = 0; i < n_nodes; ++i) out[i] = 0.0;
  for (const auto &[id, val] : res) {
    for (int64_t i = 0; i < n_nodes; ++i) {
      if ((uint64_t)nodes[i] == id) {
        out[i] = val;
        break;
      }
    }
  }
}

std::vector<std::pair<uint64_t, uint64_t>> pairs_of(const int64_t *e, int64_t n) {
  std::vector<std::pair<uint64_t, uint64_t>> v;
  v.reserve(n);
  for (int64_t i = 0; i < n; ++i) v.emplace_back((uint64_t)e[2 * i], (uint64_t)e[2 * i + 1]);
  return v;
}

}  // namespace

extern "C" {

// ---- pagerank_online (distribution-level pin only) -----------------------

void ref_pron_reset() { pagerank_online_alg::Reset(); }

void ref_pron_set(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                  const int64_t *src, const int64_t *dst, int64_t R, double eps,
                  double *out_rank) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       mg_graph::GraphType::kDirectedGraph);
  auto res = pagerank_online_alg::SetPagerank(*g, (uint64_t)R, eps);
  fill_rank(res, n_nodes, nodes, out_rank);
}

void ref_pron_update(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                     const int64_t *src, const int64_t *dst, const int64_t *created_v,
                     int64_t n_cv, const int64_t *created_e, int64_t n_ce,
                     const int64_t *deleted_v, int64_t n_dv, const int64_t *deleted_e,
                     int64_t n_de, double *out_rank) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       mg_graph::GraphType::kDirectedGraph);
  std::vector<uint64_t> cv(created_v, created_v + n_cv);
  std::vector<uint64_t> dv(deleted_v, deleted_v + n_dv);
  auto res = pagerank_online_alg::UpdatePagerank(*g, cv, pairs_of(created_e, n_ce), dv,
                                                 pairs_of(deleted_e, n_de));
  fill_rank(res, n_nodes, nodes, out_rank);
}

// ---- katz online (deterministic; exact pin) ------------------------------

void ref_katz_online_reset() { katz_alg::Reset(); }

void ref_katz_online_set(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                         const int64_t *src, const int64_t *dst, double alpha,
                         double epsilon, double *out_centrality) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       mg_graph::GraphType::kDirectedGraph);
  auto res = katz_alg::SetKatz(*g, alpha, epsilon);
  fill_rank(res, n_nodes, nodes, out_centrality);
}

// created-edge inner ids: the module passes GetInnerEdgeId of each created
// relationship (katz_centrality_online_module.cpp:110-114). mg_graph assigns
// inner edge ids in CreateEdge call order, so the caller passes the INDEX of
// each created edge within the (src,dst) arrays.
void ref_katz_online_update(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                            const int64_t *src, const int64_t *dst,
                            const int64_t *created_v, int64_t n_cv,
                            const int64_t *created_e, int64_t n_ce,
                            const int64_t *created_e_idx /* [n_ce] edge indices */,
                            const int64_t *deleted_v, int64_t n_dv,
                            const int64_t *deleted_e, int64_t n_de,
                            double *out_centrality) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       mg_graph::GraphType::kDirectedGraph);
  std::vector<uint64_t> cv(created_v, created_v + n_cv);
  std::vector<uint64_t> dv(deleted_v, deleted_v + n_dv);
  std::vector<uint64_t> ceid(created_e_idx, created_e_idx + n_ce);
  auto res = katz_alg::UpdateKatz(*g, cv, pairs_of(created_e, n_ce), ceid, dv,
                                  pairs_of(deleted_e, n_de));
  fill_rank(res, n_nodes, nodes, out_centrality);
}

}  // extern "C"

// ---- LabelRankT (community_detection_online; deterministic) --------------

namespace {
std::unique_ptr<LabelRankT::LabelRankT> lrt_instance =
    std::make_unique<LabelRankT::LabelRankT>();
}

extern "C" {

void ref_lrt_reset() { lrt_instance = std::make_unique<LabelRankT::LabelRankT>(); }

void ref_lrt_set(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                 const int64_t *src, const int64_t *dst, const double *weights,
                 int32_t directed, int32_t weighted, double similarity_threshold,
                 double exponent, double min_value, double w_selfloop,
                 int64_t max_iterations, int64_t max_updates, int64_t *out_labels) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       directed ? mg_graph::GraphType::kDirectedGraph
                                : mg_graph::GraphType::kUndirectedGraph,
                       weights);
  auto labels = lrt_instance->SetLabels(std::move(g), directed != 0, weighted != 0,
                                       similarity_threshold, exponent, min_value,
                                       "weight", w_selfloop, (uint64_t)max_iterations,
                                       (uint64_t)max_updates);
  for (int64_t i = 0; i < n_nodes; ++i) {
    auto it = labels.find((uint64_t)nodes[i]);
    out_labels[i] = it == labels.end() ? -1 : it->second;
  }
}

void ref_lrt_get(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                 const int64_t *src, const int64_t *dst, const double *weights,
                 int32_t directed, int64_t *out_labels) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       directed ? mg_graph::GraphType::kDirectedGraph
                                : mg_graph::GraphType::kUndirectedGraph,
                       weights);
  auto labels = lrt_instance->GetLabels(std::move(g));
  for (int64_t i = 0; i < n_nodes; ++i) {
    auto it = labels.find((uint64_t)nodes[i]);
    out_labels[i] = it == labels.end() ? -1 : it->second;
  }
}

void ref_lrt_update(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                    const int64_t *src, const int64_t *dst, const double *weights,
                    int32_t directed, const int64_t *mod_v, int64_t n_mv,
                    const int64_t *mod_e, int64_t n_me, const int64_t *del_v,
                    int64_t n_dv, const int64_t *del_e, int64_t n_de,
                    int64_t *out_labels) {
  auto g = build_graph(n_nodes, nodes, n_edges, src, dst,
                       directed ? mg_graph::GraphType::kDirectedGraph
                                : mg_graph::GraphType::kUndirectedGraph,
                       weights);
  std::vector<uint64_t> mv(mod_v, mod_v + n_mv);
  std::vector<uint64_t> dv(del_v, del_v + n_dv);
  std::vector<std::pair<uint64_t, uint64_t>> me, de;
  for (int64_t i = 0; i < n_me; ++i)
    me.emplace_back((uint64_t)mod_e[2 * i], (uint64_t)mod_e[2 * i + 1]);
  for (int64_t i = 0; i < n_de; ++i)
    de.emplace_back((uint64_t)del_e[2 * i], (uint64_t)del_e[2 * i + 1]);
  auto labels = lrt_instance->UpdateLabels(std::move(g), mv, me, dv, de);
  for (int64_t i = 0; i < n_nodes; ++i) {
    auto it = labels.find((uint64_t)nodes[i]);
    out_labels[i] = it == labels.end() ? -1 : it->second;
  }
}

}  // extern "C"

// ---- Leiden (randomized: distribution-level pin; goldens are stable) -----
// Drives leiden_alg::GetCommunities (src/mage/cpp/
// leiden_community_detection_module/algorithm/leiden.cpp:569-591) compiled
// from the reference sources (boost replaced by the std shim in shim/boost).
#include "leiden.hpp"

extern "C" void ref_leiden(int64_t n_nodes, int64_t n_edges, const int64_t *src,
                           const int64_t *dst, const double *weights, double gamma,
                           double theta, double resolution, int64_t max_iterations,
                           int64_t cap, int64_t *out_hier /* [n_nodes*cap], -1 pad */,
                           int64_t *out_levels /* [n_nodes] */) {
  mg_graph::Graph<> g;
  for (int64_t v = 0; v < n_nodes; ++v) g.CreateNode((uint64_t)v);
  for (int64_t e = 0; e < n_edges; ++e) {
    g.CreateEdge((uint64_t)src[e], (uint64_t)dst[e], mg_graph::GraphType::kUndirectedGraph,
                 std::nullopt, weights != nullptr, weights ? weights[e] : 0.0);
  }
  std::vector<std::vector<uint64_t>> hier;
  try {
    hier = leiden_alg::GetCommunities(g, gamma, theta, resolution,
                                      (uint64_t)max_iterations);
  } catch (const std::exception &) {
    // the reference throws "No communities detected." when its (randomized)
    // trajectory finishes without an aggregation level (leiden.cpp:585-586)
    // — surface as levels = -1 so callers can treat the run as a
    // reference-error outcome
    for (int64_t v = 0; v < n_nodes; ++v) {
      out_levels[v] = -1;
      for (int64_t k = 0; k < cap; ++k) out_hier[v * cap + k] = -1;
    }
    return;
  }
  for (int64_t v = 0; v < n_nodes; ++v) {
    int64_t L = 0;
    if ((size_t)v < hier.size()) {
      for (auto c : hier[v]) {
        if (L < cap) out_hier[v * cap + L] = (int64_t)c;
        ++L;
      }
    }
    for (int64_t k = L; k < cap; ++k) out_hier[v * cap + k] = -1;
    out_levels[v] = L;
  }
}
