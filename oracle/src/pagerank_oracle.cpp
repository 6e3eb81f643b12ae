// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
// Restates pagerank_alg::ParallelIterativePageRank,
// /root/reference/src/mage/cpp/pagerank_module/algorithm/pagerank.cpp:194-242.

#include <algorithm>
#include <chrono>
#include <cmath>
#include <cstring>
#include <thread>
#include <vector>

#include "../oracle.h"

namespace {

// One power iteration, fp64, pull style (pagerank.cpp:87-97 + :105-113 +
// :223-224): new[v] = (1-d)/N + d * sum_{u->v} rank[u]/outdeg(u).
void Iterate(int64_t n_vertices, int64_t n_edges, const int64_t *src, const int64_t *dst,
             const std::vector<double> &out_degree, double damping, const std::vector<double> &rank,
             std::vector<double> &next) {
  const double base = (1.0 - damping) / static_cast<double>(n_vertices);
  std::fill(next.begin(), next.end(), 0.0);
  for (int64_t e = 0; e < n_edges; ++e) {
    next[dst[e]] += rank[src[e]] / out_degree[src[e]];
  }
  for (int64_t v = 0; v < n_vertices; ++v) {
    next[v] = base + damping * next[v];
  }
}

// pagerank.cpp:139-151: continue while any |new-old| > stop_epsilon and
// iteration count below max.
bool ContinueIterate(const std::vector<double> &a, const std::vector<double> &b,
                     int64_t iters, int64_t max_iters, double eps) {
  if (iters == max_iters) return false;
  for (size_t i = 0; i < a.size(); ++i) {
    if (std::abs(a[i] - b[i]) > eps) return true;
  }
  return false;
}

}  // namespace

extern "C" int64_t oracle_pagerank(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                   const int64_t *dst, int64_t max_iterations,
                                   double damping_factor, double stop_epsilon,
                                   double *out_rank) {
  if (n_vertices < 0 || n_edges < 0) return -1;
  if (n_vertices == 0) return 0;

  std::vector<double> out_degree(n_vertices, 0.0);
  for (int64_t e = 0; e < n_edges; ++e) {
    if (src[e] < 0 || src[e] >= n_vertices || dst[e] < 0 || dst[e] >= n_vertices) return -1;
    out_degree[src[e]] += 1.0;
  }

  std::vector<double> rank(n_vertices, 1.0 / static_cast<double>(n_vertices));
  std::vector<double> next(n_vertices);

  // Loop structure of pagerank.cpp:218-239: iterate, swap, then test on the
  // (new, old) pair.
  int64_t iters = 0;
  bool cont = max_iterations != 0;
  while (cont) {
    Iterate(n_vertices, n_edges, src, dst, out_degree, damping_factor, rank, next);
    rank.swap(next);
    ++iters;
    cont = ContinueIterate(rank, next, iters, max_iterations, stop_epsilon);
  }

  // NormalizeRank, pagerank.cpp:157-162.
  double sum = 0.0;
  for (double v : rank) sum += v;
  for (int64_t v = 0; v < n_vertices; ++v) out_rank[v] = rank[v] / sum;
  return iters;
}

extern "C" double oracle_pagerank_timed(int64_t n_vertices, int64_t n_edges, const int64_t *src,
                                        const int64_t *dst, int64_t iterations,
                                        double damping_factor, int64_t n_threads,
                                        double *out_rank) {
  // Multithreaded variant with the reference's edge-block decomposition
  // (pagerank.cpp:61-72 CalculateOptimalBorders + :215-235 per-thread blocks
  // merged into rank_next) for bench.py's cpu_baseline. Timing covers the
  // iteration loop only.
  if (n_vertices <= 0) return 0.0;
  if (n_threads < 1) n_threads = 1;

  std::vector<double> out_degree(n_vertices, 0.0);
  for (int64_t e = 0; e < n_edges; ++e) out_degree[src[e]] += 1.0;

  std::vector<double> rank(n_vertices, 1.0 / static_cast<double>(n_vertices));
  const double base = (1.0 - damping_factor) / static_cast<double>(n_vertices);

  std::vector<int64_t> borders(n_threads + 1);
  for (int64_t b = 0; b <= n_threads; ++b) borders[b] = b * n_edges / n_threads;

  std::vector<std::vector<double>> blocks(n_threads);

  auto t0 = std::chrono::steady_clock::now();
  for (int64_t it = 0; it < iterations; ++it) {
    std::vector<std::thread> threads;
    threads.reserve(n_threads);
    for (int64_t t = 0; t < n_threads; ++t) {
      threads.emplace_back([&, t] {
        std::vector<double> block(n_vertices, 0.0);
        for (int64_t e = borders[t]; e < borders[t + 1]; ++e) {
          block[dst[e]] += rank[src[e]] / out_degree[src[e]];
        }
        blocks[t] = std::move(block);
      });
    }
    for (auto &th : threads) th.join();
    std::vector<double> next(n_vertices, base);
    for (int64_t t = 0; t < n_threads; ++t) {
      for (int64_t v = 0; v < n_vertices; ++v) next[v] += damping_factor * blocks[t][v];
    }
    rank.swap(next);
  }
  auto t1 = std::chrono::steady_clock::now();

  double sum = 0.0;
  for (double v : rank) sum += v;
  if (out_rank) {
    for (int64_t v = 0; v < n_vertices; ++v) out_rank[v] = rank[v] / sum;
  }
  return std::chrono::duration<double>(t1 - t0).count();
}
