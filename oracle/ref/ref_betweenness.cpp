// ORACLE/_REF — TEST INFRASTRUCTURE ONLY (see ref_wrap.cpp header).
// Drives the reference Brandes core (betweenness_centrality.cpp) through a
// header-only mg_graph, as the module's GetGraphView would build it.

#include <cstdint>
#include <vector>

#include <mg_graph.hpp>

#include "betweenness_centrality.hpp"

extern "C" int64_t ref_betweenness(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                   const int64_t *dst, int32_t directed, int32_t normalize,
                                   int64_t n_threads, double *out_bc) {
  mg_graph::Graph<> g;
  for (int64_t v = 0; v < n_vertices; ++v) g.CreateNode((uint64_t)v);
  const auto type =
      directed ? mg_graph::GraphType::kDirectedGraph : mg_graph::GraphType::kUndirectedGraph;
  for (int64_t e = 0; e < n_edges; ++e) {
    g.CreateEdge((uint64_t)src[e], (uint64_t)dst[e], type);
  }
  auto bc = betweenness_centrality_alg::BetweennessCentrality(g, directed != 0,
                                                              normalize != 0,
                                                              (int)n_threads);
  for (size_t i = 0; i < bc.size(); ++i) out_bc[i] = bc[i];
  return 0;
}
