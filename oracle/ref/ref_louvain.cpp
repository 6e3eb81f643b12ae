// ORACLE/_REF — TEST INFRASTRUCTURE ONLY (see ref_wrap.cpp header).
// Drives the reference grappolo basic Louvain path. The COO→symmetrized-CSR
// step restates GetGrappoloSuitableGraph (louvain.cpp:158-233) because the
// module wrapper louvain.cpp is not compilable here (needs <format> via
// mgp.hpp); the multiphase algorithm itself is the REFERENCE's own code.

#include <cstdint>
#include <cstdlib>
#include <vector>

#include "defs.h"  // grappolo/DefineStructure/defs.h

// grappolo/BasicCommunitiesDetection/basic_comm.h signature (mgp_graph is an
// opaque pointer here; only the allocation-tracking stubs receive it).
struct mgp_graph;
extern void runMultiPhaseBasic(graph *G, mgp_graph *mg_graph, long *C_orig, int basicOpt,
                               long minGraphSize, double threshold, double C_threshold,
                               int numThreads, int threadsOpt);

extern "C" int64_t ref_louvain(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                               const int64_t *dst, const double *weights, double threshold,
                               int64_t n_threads, int64_t *out_community) {
  if (n_vertices == 0 || n_edges == 0) return 0;

  // Symmetrized CSR, each edge stored twice (louvain.cpp:176-233 shape).
  long *ptrs = (long *)calloc(n_vertices + 1, sizeof(long));
  edge *list = (edge *)malloc(2 * n_edges * sizeof(edge));
  for (int64_t e = 0; e < n_edges; ++e) {
    ++ptrs[src[e] + 1];
    ++ptrs[dst[e] + 1];
  }
  for (int64_t v = 0; v < n_vertices; ++v) ptrs[v + 1] += ptrs[v];
  std::vector<long> cur(ptrs, ptrs + n_vertices);
  for (int64_t e = 0; e < n_edges; ++e) {
    const double w = weights ? weights[e] : 1.0;
    list[cur[src[e]]++] = {(long)src[e], (long)dst[e], w};
    list[cur[dst[e]]++] = {(long)dst[e], (long)src[e], w};
  }

  graph *G = (graph *)malloc(sizeof(graph));
  G->numVertices = n_vertices;
  G->sVertices = n_vertices;
  G->numEdges = n_edges;
  G->edgeListPtrs = ptrs;
  G->edgeList = list;

  std::vector<long> C(n_vertices, -1);
  // Argument values as the module passes them (community_detection_module
  // wrapper: kReplaceMap=0, min_graph_shrink, coloring_threshold unused on
  // the basic path, kThreadsOpt=1).
  runMultiPhaseBasic(G, nullptr, C.data(), /*basicOpt=*/0, /*minGraphSize=*/100000, threshold,
                     /*C_threshold=*/0.01, (int)n_threads, /*threadsOpt=*/1);
  // runMultiPhaseBasic frees G and its arrays.

  int64_t max_c = -1;
  for (int64_t i = 0; i < n_vertices; ++i) {
    out_community[i] = C[i];
    if (C[i] > max_c) max_c = C[i];
  }
  return max_c + 1;
}
