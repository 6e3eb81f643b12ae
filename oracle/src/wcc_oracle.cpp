// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
// Restates Weak,
// /root/reference/src/mage/cpp/connectivity_module/connectivity_module.cpp:41-87.

#include <cstdint>
#include <queue>
#include <vector>

#include "../oracle.h"

extern "C" int64_t oracle_wcc(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                              const int64_t *dst, int64_t *out_component) {
  if (n_vertices < 0 || n_edges < 0) return -1;

  // Undirected adjacency, the shape GetGraphView(kUndirectedGraph) builds
  // (include/mg_graph.hpp:229-231 pushes each edge both ways).
  std::vector<int64_t> degree(n_vertices, 0);
  for (int64_t e = 0; e < n_edges; ++e) {
    if (src[e] < 0 || src[e] >= n_vertices || dst[e] < 0 || dst[e] >= n_vertices) return -1;
    ++degree[src[e]];
    ++degree[dst[e]];
  }
  std::vector<int64_t> offset(n_vertices + 1, 0);
  for (int64_t v = 0; v < n_vertices; ++v) offset[v + 1] = offset[v] + degree[v];
  std::vector<int64_t> adj(offset[n_vertices]);
  std::vector<int64_t> cursor(offset.begin(), offset.end() - 1);
  for (int64_t e = 0; e < n_edges; ++e) {
    adj[cursor[src[e]]++] = dst[e];
    adj[cursor[dst[e]]++] = src[e];
  }

  // BFS from each unvisited vertex in scan order; component ids in
  // root-discovery order (connectivity_module.cpp:47-72).
  const int64_t kUnset = -1;
  for (int64_t v = 0; v < n_vertices; ++v) out_component[v] = kUnset;
  int64_t curr_component = 0;
  std::queue<int64_t> q;
  for (int64_t v = 0; v < n_vertices; ++v) {
    if (out_component[v] != kUnset) continue;
    out_component[v] = curr_component;
    q.push(v);
    while (!q.empty()) {
      int64_t u = q.front();
      q.pop();
      for (int64_t j = offset[u]; j < offset[u + 1]; ++j) {
        int64_t w = adj[j];
        if (out_component[w] != kUnset) continue;
        out_component[w] = curr_component;
        q.push(w);
      }
    }
    ++curr_component;
  }
  return curr_component;
}
