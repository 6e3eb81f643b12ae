// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
// Restates katz_alg::SetKatz / KatzCentralityLoop / Converged,
// /root/reference/src/mage/cpp/katz_centrality_module/algorithm/katz.cpp
// :393-414 (SetKatz), :226-255 (loop), :165-215 (Converged).

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <numeric>
#include <vector>

#include "../oracle.h"

extern "C" int64_t oracle_katz(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                               const int64_t *dst, double alpha, double epsilon,
                               double *out_centrality) {
  if (n_vertices < 0 || n_edges < 0) return -1;

  for (int64_t v = 0; v < n_vertices; ++v) out_centrality[v] = 0.0;
  // SetKatz returns the all-zero init when the graph has no edges
  // (katz.cpp:398-400: context.Init + early WrapResults).
  if (n_edges == 0) return 0;

  std::vector<int64_t> out_degree(n_vertices, 0);
  for (int64_t e = 0; e < n_edges; ++e) {
    if (src[e] < 0 || src[e] >= n_vertices || dst[e] < 0 || dst[e] >= n_vertices) return -1;
    ++out_degree[src[e]];
  }

  // MaxDegree (katz.cpp:137-148) over Neighbours() of a directed GraphView,
  // which are the OUT-neighbours (include/mg_graph.hpp:96-109).
  int64_t deg_max = 0;
  for (int64_t v = 0; v < n_vertices; ++v) deg_max = std::max(deg_max, out_degree[v]);
  // gamma, katz.cpp:403-404 (IEEE semantics kept: may be inf/negative for
  // alpha^2*degmax >= 1 — the reference computes exactly this).
  const double gamma =
      static_cast<double>(deg_max) / (1.0 - (alpha * alpha * static_cast<double>(deg_max)));

  // Dense state. omega_0 = 1 (Init, katz.cpp:41-44); centrality_0 = 0.
  std::vector<double> omega_prev(n_vertices, 1.0), omega(n_vertices);
  std::vector<double> centrality(n_vertices, 0.0), lr(n_vertices, 0.0), ur(n_vertices, 0.0);

  int64_t iteration = 0;
  std::vector<int64_t> order(n_vertices);
  while (true) {
    ++iteration;
    // omega_i(v) = sum over in-neighbours u of omega_{i-1}(u)
    // (katz.cpp:238-241 walks InNeighbours).
    std::fill(omega.begin(), omega.end(), 0.0);
    for (int64_t e = 0; e < n_edges; ++e) omega[dst[e]] += omega_prev[src[e]];
    const double a_i = std::pow(alpha, static_cast<double>(iteration));
    const double a_i1 = std::pow(alpha, static_cast<double>(iteration + 1));
    for (int64_t v = 0; v < n_vertices; ++v) {
      centrality[v] += a_i * omega[v];
      lr[v] = centrality[v];                         // katz.cpp:247
      ur[v] = centrality[v] + a_i1 * omega[v] * gamma;  // katz.cpp:248-250
    }
    omega_prev.swap(omega);

    // Converged (katz.cpp:165-215) AFTER the k = centrality.size() override
    // at :172: the deactivation loop (:191-197) and the size>k test (:201)
    // are dead; what remains is: sort all centralities descending, converged
    // iff ur(v_i) - epsilon < lr(v_{i-1}) for every adjacent sorted pair
    // (:206-213). Tie order in the reference's partial_sort is unspecified;
    // we break ties by node id ascending (documented divergence).
    std::iota(order.begin(), order.end(), 0);
    std::sort(order.begin(), order.end(), [&](int64_t a, int64_t b) {
      if (centrality[a] != centrality[b]) return centrality[a] > centrality[b];
      return a < b;
    });
    bool converged = true;
    for (int64_t j = 1; j < n_vertices; ++j) {
      if (ur[order[j]] - epsilon >= lr[order[j - 1]]) {
        converged = false;
        break;
      }
    }
    if (converged) break;
    if (iteration > 1000000) return -1;  // safety net, not in reference
  }

  for (int64_t v = 0; v < n_vertices; ++v) out_centrality[v] = centrality[v];
  return iteration;
}
