// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header note).
//
// Seeded restatement of pagerank_online_alg (/root/reference/query_modules/
// pagerank_module/algorithm_online/pagerank.cpp): R random walks per node
// (SetPagerank :253-278), epsilon-stop routes (CreateRoute :110-134),
// truncate-after-first-occurrence + eps/2 regrow updates (UpdateCreate
// :143-188, UpdateDelete :197-239), rank = visits/sum (CalculatePageRank
// :83-98 — the ((n*R)/eps) scaling cancels in NormalizeRank :70-76).
//
// The reference seeds two static std::minstd_rand engines from
// std::random_device (:53-63); parity with it is therefore STATISTICAL
// (DESIGN.md "Statistical-parity bar"). This restatement takes an explicit
// seed so test distributions are reproducible. Replicated faithfully
// otherwise, with one divergence: walks_counter is signed and clamped at 0
// (the reference's uint64 map can underflow on decrement of an erased
// entry — a latent bug we do not copy).

#include <algorithm>
#include <cstdint>
#include <random>
#include <set>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace {

struct Ctx {
  std::vector<std::vector<int64_t>> walks;
  std::unordered_map<int64_t, int64_t> walks_counter;
  std::unordered_map<int64_t, std::unordered_set<int64_t>> walks_table;
  std::minstd_rand eng_int, eng_float;
  int64_t R = 10;
  double eps = 0.2;
  bool initialized = false;

  void init_engines(uint64_t seed) {
    // splitmix-style spreading so nearby seeds give unrelated streams
    uint64_t x = seed + 0x9E3779B97F4A7C15ull;
    x = (x ^ (x >> 30)) * 0xBF58476D1CE4E5B9ull;
    x = (x ^ (x >> 27)) * 0x94D049BB133111EBull;
    x ^= x >> 31;
    eng_int.seed((uint32_t)(x & 0xFFFFFFFFu) | 1u);
    eng_float.seed((uint32_t)(x >> 32) | 1u);
  }
  int rand_int(int from, int to) {  // [from, to-1] (pagerank.cpp:53-57)
    std::uniform_int_distribution<int> dist{from, to - 1};
    return dist(eng_int);
  }
  float rand_float() {  // [0,1) (pagerank.cpp:59-63)
    std::uniform_real_distribution<float> dist{};
    return dist(eng_float);
  }
};

Ctx g;

struct Graph {
  std::vector<int64_t> nodes;
  std::unordered_map<int64_t, std::vector<int64_t>> out;  // multi-edges kept
  std::unordered_set<int64_t> node_set;
};

Graph build(int64_t n_nodes, const int64_t *nodes, int64_t n_edges, const int64_t *src,
            const int64_t *dst) {
  Graph gr;
  gr.nodes.assign(nodes, nodes + n_nodes);
  for (int64_t i = 0; i < n_nodes; ++i) gr.node_set.insert(nodes[i]);
  for (int64_t e = 0; e < n_edges; ++e) gr.out[src[e]].push_back(dst[e]);
  return gr;
}

// CreateRoute (pagerank.cpp:110-134)
void create_route(const Graph &gr, int64_t start_id, std::vector<int64_t> &walk,
                  int64_t walk_index, double epsilon) {
  int64_t current = start_id;
  while (true) {
    auto it = gr.out.find(current);
    if (it == gr.out.end() || it->second.empty()) break;
    const auto &nb = it->second;
    int64_t next = nb[g.rand_int(0, (int)nb.size())];
    walk.push_back(next);
    g.walks_table[next].insert(walk_index);
    g.walks_counter[next]++;
    if (g.rand_float() < epsilon) break;
    current = next;
  }
}

// UpdateCreate(edge)/UpdateDelete(edge) share the truncate+regrow body
// (pagerank.cpp:143-163, 197-228).
void rewire(const Graph &gr, int64_t from, bool allow_regrow) {
  auto t = g.walks_table.find(from);
  if (t == g.walks_table.end()) return;
  std::unordered_set<int64_t> copy = t->second;
  for (int64_t wi : copy) {
    auto &walk = g.walks[wi];
    auto pos = std::find(walk.begin(), walk.end(), from);
    if (pos == walk.end()) continue;
    ++pos;
    for (auto p = pos; p != walk.end(); ++p) {
      g.walks_table[*p].erase(wi);
      auto c = g.walks_counter.find(*p);
      if (c != g.walks_counter.end() && c->second > 0) c->second--;
    }
    walk.erase(pos, walk.end());
    if (!allow_regrow) continue;
    if (!gr.node_set.count(from)) continue;  // NodeExists (:221-223)
    create_route(gr, from, walk, wi, g.eps / 2.0);
  }
}

void spawn_walks(const Graph &gr, int64_t v) {  // UpdateCreate(vertex) :172-188
  int64_t wi = (int64_t)g.walks.size();
  for (int64_t r = 0; r < g.R; ++r) {
    std::vector<int64_t> walk{v};
    g.walks_table[v].insert(wi);
    g.walks_counter[v]++;
    create_route(gr, v, walk, wi, g.eps);
    g.walks.push_back(std::move(walk));
    ++wi;
  }
}

// CalculatePageRank + NormalizeRank (:70-98): rank by node id into out_rank
// (indexed by position in `nodes`), normalized over ALL counter entries.
void ranks_out(int64_t n_nodes, const int64_t *nodes, double *out_rank) {
  double sum = 0.0;
  for (auto &[id, c] : g.walks_counter) sum += (double)c;
  for (int64_t i = 0; i < n_nodes; ++i) {
    auto it = g.walks_counter.find(nodes[i]);
    out_rank[i] = (it != g.walks_counter.end() && sum > 0) ? (double)it->second / sum : 0.0;
  }
}

}  // namespace

extern "C" {

void oracle_pronline_reset() { g = Ctx{}; }

int oracle_pronline_initialized() { return g.initialized && !g.walks.empty() ? 1 : 0; }

// SetPagerank (:253-278)
void oracle_pronline_set(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                         const int64_t *src, const int64_t *dst, int64_t R, double eps,
                         uint64_t seed, double *out_rank) {
  g = Ctx{};
  g.init_engines(seed);
  g.R = R;
  g.eps = eps;
  g.initialized = true;
  Graph gr = build(n_nodes, nodes, n_edges, src, dst);
  for (int64_t i = 0; i < n_nodes; ++i) spawn_walks(gr, nodes[i]);
  if (out_rank) ranks_out(n_nodes, nodes, out_rank);
}

// GetPagerank consistency test (IsIncosistent :241-250). Returns 1 when
// consistent (and fills ranks), 0 when the caller must raise the error.
int oracle_pronline_get(int64_t n_nodes, const int64_t *nodes, double *out_rank) {
  for (int64_t i = 0; i < n_nodes; ++i)
    if (!g.walks_counter.count(nodes[i])) return 0;
  if (out_rank) ranks_out(n_nodes, nodes, out_rank);
  return 1;
}

// UpdatePagerank (:292-315): deleted edges, deleted vertices, created
// vertices, created edges — in that order.
void oracle_pronline_update(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                            const int64_t *src, const int64_t *dst,
                            const int64_t *created_v, int64_t n_cv,
                            const int64_t *created_e, int64_t n_ce,
                            const int64_t *deleted_v, int64_t n_dv,
                            const int64_t *deleted_e, int64_t n_de, double *out_rank) {
  Graph gr = build(n_nodes, nodes, n_edges, src, dst);
  for (int64_t i = 0; i < n_de; ++i) rewire(gr, deleted_e[2 * i], true);
  for (int64_t i = 0; i < n_dv; ++i) {
    // UpdateDelete(vertex) :236-239
    g.walks_table.erase(deleted_v[i]);
    g.walks_counter.erase(deleted_v[i]);
  }
  for (int64_t i = 0; i < n_cv; ++i) spawn_walks(gr, created_v[i]);
  for (int64_t i = 0; i < n_ce; ++i) rewire(gr, created_e[2 * i], true);
  if (out_rank) ranks_out(n_nodes, nodes, out_rank);
}

// Test support: structural invariants.
void oracle_pronline_stats(int64_t *n_walks, int64_t *total_visits) {
  if (n_walks) *n_walks = (int64_t)g.walks.size();
  if (total_visits) {
    int64_t s = 0;
    for (auto &[id, c] : g.walks_counter) s += c;
    *total_visits = s;
  }
}

}  // extern "C"
