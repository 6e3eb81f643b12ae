// ORACLE/_REF — TEST INFRASTRUCTURE ONLY (see ref_wrap.cpp header).
// Drives the reference katz core (katz.cpp SetKatz static path) through a
// header-only mg_graph::Graph, as mg_utility::GetGraphView(kDirectedGraph)
// would build it (include/mg_utils.hpp:127-150).

#include <cstdint>
#include <vector>

#include <mg_graph.hpp>

#include "katz.hpp"  // reference katz_centrality_module/algorithm/katz.hpp

extern "C" int64_t ref_katz(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                            const int64_t *dst, double alpha, double epsilon,
                            double *out_centrality) {
  mg_graph::Graph<> g;
  for (int64_t v = 0; v < n_vertices; ++v) g.CreateNode((uint64_t)v);
  for (int64_t e = 0; e < n_edges; ++e) {
    g.CreateEdge((uint64_t)src[e], (uint64_t)dst[e], mg_graph::GraphType::kDirectedGraph);
  }
  auto res = katz_alg::SetKatz(g, alpha, epsilon);
  for (const auto &[id, c] : res) out_centrality[id] = c;
  return (int64_t)res.size();
}
