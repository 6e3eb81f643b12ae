// ORACLE/_REF — TEST INFRASTRUCTURE ONLY.
// C entry points over the REFERENCE's own algorithm cores, compiled from the
// sources where they lie under /root/reference (never copied into this
// repo). Outputs go only to oracle/_ref/ (git-ignored, travels to the GPU
// box). Used to pin the oracle restatements and as bench.py's
// cpu_baseline kind="reference".
//
// Compiled units (recipe: oracle/ref/Makefile):
//   - pagerank core: src/mage/cpp/pagerank_module/algorithm/pagerank.cpp
//   - grappolo Louvain basic path (+2-symbol mgp allocation-tracking stub)
//   - katz core: src/mage/cpp/katz_centrality_module/algorithm/katz.cpp
//     (compiled against a local shim for the never-called mgp::Graph
//     online-update overloads; see shim/)
// WCC has no separable reference core (30 lines inline in the module, pinned
// by e2e goldens + the oracle restatement instead).

#include <chrono>
#include <cstdint>
#include <cstring>
#include <utility>
#include <vector>

#include "pagerank.hpp"  // reference pagerank_module/algorithm/pagerank.hpp

extern "C" int64_t ref_pagerank(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                const int64_t *dst, int64_t max_iterations, double damping,
                                double stop_epsilon, int64_t n_threads, double *out_rank) {
  std::vector<pagerank_alg::EdgePair> edges;
  edges.reserve(n_edges);
  for (int64_t e = 0; e < n_edges; ++e) {
    edges.emplace_back((uint64_t)src[e], (uint64_t)dst[e]);
  }
  pagerank_alg::PageRankGraph graph((uint64_t)n_vertices, (uint64_t)n_edges, edges);
  auto rank = pagerank_alg::ParallelIterativePageRank(graph, (size_t)max_iterations, damping,
                                                      stop_epsilon, (uint32_t)n_threads);
  for (size_t i = 0; i < rank.size(); ++i) out_rank[i] = rank[i];
  return (int64_t)rank.size();
}

extern "C" double ref_pagerank_timed(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                     const int64_t *dst, int64_t iterations, double damping,
                                     int64_t n_threads, double *out_rank) {
  std::vector<pagerank_alg::EdgePair> edges;
  edges.reserve(n_edges);
  for (int64_t e = 0; e < n_edges; ++e) {
    edges.emplace_back((uint64_t)src[e], (uint64_t)dst[e]);
  }
  pagerank_alg::PageRankGraph graph((uint64_t)n_vertices, (uint64_t)n_edges, edges);
  auto t0 = std::chrono::steady_clock::now();
  auto rank = pagerank_alg::ParallelIterativePageRank(graph, (size_t)iterations, damping,
                                                      /*stop_epsilon=*/0.0, (uint32_t)n_threads);
  auto t1 = std::chrono::steady_clock::now();
  if (out_rank) {
    for (size_t i = 0; i < rank.size(); ++i) out_rank[i] = rank[i];
  }
  return std::chrono::duration<double>(t1 - t0).count();
}
