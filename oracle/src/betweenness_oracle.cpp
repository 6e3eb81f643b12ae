// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
// Restates betweenness_centrality_alg::BetweennessCentrality (sequential),
// /root/reference/src/mage/cpp/betweenness_centrality_module/algorithm/
// betweenness_centrality.cpp:25-54 (BFS with predecessor lists and uint64
// path counters) and :71-147 (dependency accumulation, /2 for undirected,
// normalization by 1/((n-1)(n-2)) directed or 2/((n-1)(n-2)) undirected,
// constant 1.0 when n <= 2).
//
// directed != 0: BFS over OUT-neighbours (mg_graph Neighbours of a directed
// view); directed == 0: over the symmetrized adjacency (both directions,
// multi-edges kept — multiplicity affects path counts exactly as the
// reference's duplicated adjacency entries do).

#include <cstdint>
#include <queue>
#include <stack>
#include <vector>

#include "../oracle.h"

extern "C" int64_t oracle_betweenness(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                      const int64_t *dst, int32_t directed, int32_t normalize,
                                      double *out_bc /* [n_vertices] */) {
  if (n_vertices < 0 || n_edges < 0) return -1;
  const int64_t V = n_vertices;
  for (int64_t v = 0; v < V; ++v) out_bc[v] = 0.0;
  if (V == 0) return 0;

  // adjacency (out-edges for directed; both ways for undirected)
  std::vector<int64_t> degree(V, 0);
  for (int64_t e = 0; e < n_edges; ++e) {
    if (src[e] < 0 || src[e] >= V || dst[e] < 0 || dst[e] >= V) return -1;
    ++degree[src[e]];
    if (!directed) ++degree[dst[e]];
  }
  std::vector<int64_t> offset(V + 1, 0);
  for (int64_t v = 0; v < V; ++v) offset[v + 1] = offset[v] + degree[v];
  std::vector<int64_t> adj(offset[V]);
  std::vector<int64_t> cur(offset.begin(), offset.end() - 1);
  for (int64_t e = 0; e < n_edges; ++e) {
    adj[cur[src[e]]++] = dst[e];
    if (!directed) adj[cur[dst[e]]++] = src[e];
  }

  std::vector<uint64_t> sigma(V);
  std::vector<int> dist(V);
  std::vector<std::vector<int64_t>> preds(V);
  std::vector<double> dep(V);

  for (int64_t s = 0; s < V; ++s) {
    std::fill(sigma.begin(), sigma.end(), 0);
    std::fill(dist.begin(), dist.end(), -1);
    std::fill(dep.begin(), dep.end(), 0.0);
    for (auto &p : preds) p.clear();

    // BFS (betweenness_centrality.cpp:25-54)
    sigma[s] = 1;
    dist[s] = 0;
    std::stack<int64_t> visited;
    std::queue<int64_t> q;
    q.push(s);
    while (!q.empty()) {
      const int64_t u = q.front();
      q.pop();
      visited.push(u);
      for (int64_t j = offset[u]; j < offset[u + 1]; ++j) {
        const int64_t w = adj[j];
        if (dist[w] < 0) {
          q.push(w);
          dist[w] = dist[u] + 1;
        }
        if (dist[w] == dist[u] + 1) {
          sigma[w] += sigma[u];  // uint64 wrap semantics kept
          preds[w].push_back(u);
        }
      }
    }

    // dependency accumulation (betweenness_centrality.cpp:92-106)
    while (!visited.empty()) {
      const int64_t w = visited.top();
      visited.pop();
      for (const int64_t p : preds[w]) {
        dep[p] += ((double)sigma[p] / (double)sigma[w]) * (1.0 + dep[w]);
      }
      if (w != s) out_bc[w] += directed ? dep[w] : dep[w] / 2.0;
    }
  }

  if (normalize) {
    // betweenness_centrality.cpp:139-144
    const double pairs = (double)((V - 1) * (V - 2));
    const double numerator = directed ? 1.0 : 2.0;
    const double constant = V > 2 ? numerator / pairs : 1.0;
    for (int64_t v = 0; v < V; ++v) out_bc[v] *= constant;
  }
  return 0;
}
