// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header note).
//
// Sequential restatement of the reference's ONLINE Katz centrality
// (/root/reference/query_modules/katz_centrality_module/algorithm/katz.cpp):
//   SetKatz (:356-378): omega_i = A^T omega_{i-1}; c_i = c_{i-1} + a^i w_i;
//     lr/ur bounds with gamma = degmax/(1 - a^2 degmax); loop until the
//     active set is epsilon-separated (Converged :139-186 — k is overridden
//     to centrality.size() at :146, which kills the erase loop :163-168 and
//     reduces the test to adjacent-pair separation of the active
//     centralities sorted descending).
//   UpdateKatz (:380-468): per-iteration omega deltas propagated along
//     CURRENT out-edges from the updated-node closure (UpdateLevel
//     :253-313, skipping created-edge instances by edge id), new-edge adds,
//     deleted-edge subtracts against the OLD omegas, centralities for the
//     updated set (i==1 special case :308-311), lr/ur refresh WITHOUT gamma
//     (:437-439), re-activation by ur >= min_lr - eps (:442-455), deleted-
//     vertex erasure, then the continuation loop with gamma =
//     degmax/(1 - a*degmax) (:396 — note NOT a^2; replicated).
// Deterministic throughout; the only divergence is the descending sort's
// tie order (std::partial_sort unspecified; here value desc, id asc).
//
// Edge ids: mg_graph assigns inner edge ids in creation order, so
// created-edge ids arrive as INDICES into the (src,dst) arrays.

#include <algorithm>
#include <cmath>
#include <cstdint>
#include <map>
#include <queue>
#include <set>
#include <unordered_map>
#include <unordered_set>
#include <vector>

namespace {

using Map = std::unordered_map<int64_t, double>;

struct KCtx {
  std::vector<Map> centralities, omegas;
  Map lr, ur;
  std::set<int64_t> active;
  int64_t iteration = 0;
  double alpha = 0.2, eps = 0.01;
  bool initialized = false;
};

KCtx g;

struct Graph {
  std::vector<int64_t> nodes;
  std::unordered_set<int64_t> node_set;
  // adjacency with edge ids (= index into the input arrays)
  std::unordered_map<int64_t, std::vector<std::pair<int64_t, int64_t>>> out, in;
  int64_t n_edges = 0;
};

Graph build(int64_t n_nodes, const int64_t *nodes, int64_t n_edges, const int64_t *src,
            const int64_t *dst) {
  Graph gr;
  gr.nodes.assign(nodes, nodes + n_nodes);
  gr.node_set.insert(nodes, nodes + n_nodes);
  gr.n_edges = n_edges;
  for (int64_t e = 0; e < n_edges; ++e) {
    gr.out[src[e]].emplace_back(dst[e], e);
    gr.in[dst[e]].emplace_back(src[e], e);
  }
  return gr;
}

void init_ctx(const Graph &gr) {
  g.centralities.clear();
  g.omegas.clear();
  g.active.clear();
  g.iteration = 0;
  Map c0, w0;
  for (auto v : gr.nodes) {
    c0[v] = 0.0;
    w0[v] = 1.0;
    g.lr[v] = 0.0;
    g.ur[v] = 0.0;
  }
  g.lr.clear();
  g.ur.clear();
  for (auto v : gr.nodes) {
    g.lr[v] = 0.0;
    g.ur[v] = 0.0;
  }
  g.centralities.push_back(std::move(c0));
  g.omegas.push_back(std::move(w0));
}

void add_iteration(const Graph &gr) {
  ++g.iteration;
  Map c, w;
  for (auto v : gr.nodes) {
    c[v] = 0.0;
    w[v] = 0.0;
  }
  g.centralities.push_back(std::move(c));
  g.omegas.push_back(std::move(w));
}

int64_t max_degree(const Graph &gr) {
  int64_t m = 0;
  for (auto v : gr.nodes) {
    auto it = gr.out.find(v);
    int64_t d = it == gr.out.end() ? 0 : (int64_t)it->second.size();
    if (d > m) m = d;
  }
  return m;
}

// Converged (:139-186) after the k-override.
bool converged() {
  const Map &cent = g.centralities[g.iteration];
  std::vector<std::pair<int64_t, double>> ac;
  for (auto v : g.active) ac.emplace_back(v, cent.at(v));
  std::sort(ac.begin(), ac.end(), [](const auto &a, const auto &b) {
    if (a.second != b.second) return a.second > b.second;
    return a.first < b.first;  // documented tie divergence
  });
  for (size_t i = 1; i < ac.size(); ++i) {
    if (g.ur.at(ac[i].first) - g.eps >= g.lr.at(ac[i - 1].first)) return false;
  }
  return true;
}

// KatzCentralityLoop (:211-240)
void katz_loop(const Graph &gr, double gamma) {
  do {
    add_iteration(gr);
    const int64_t i = g.iteration;
    for (auto v : gr.nodes) {
      double acc = 0.0;
      auto it = gr.in.find(v);
      if (it != gr.in.end())
        for (auto [u, eid] : it->second) acc += g.omegas[i - 1].at(u);
      g.omegas[i][v] = acc;
      g.centralities[i][v] = g.centralities[i - 1].at(v) + pow(g.alpha, (double)i) * acc;
      g.lr[v] = g.centralities[i][v];
      g.ur[v] = g.centralities[i][v] + pow(g.alpha, (double)(i + 1)) * acc * gamma;
    }
  } while (!converged());
}

}  // namespace

extern "C" {

void oracle_konline_reset() { g = KCtx{}; }

int oracle_konline_initialized() {
  return (g.initialized && !g.centralities.empty() && g.iteration > 0) ? 1 : 0;
}

int64_t oracle_konline_iterations() { return g.iteration; }

void oracle_konline_set(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                        const int64_t *src, const int64_t *dst, double alpha, double eps,
                        double *out /* [n_nodes] by position */) {
  g = KCtx{};
  g.alpha = alpha;
  g.eps = eps;
  g.initialized = true;
  Graph gr = build(n_nodes, nodes, n_edges, src, dst);
  init_ctx(gr);
  if (n_edges == 0) {
    for (int64_t i = 0; i < n_nodes; ++i) out[i] = 0.0;
    return;
  }
  const double dm = (double)max_degree(gr);
  const double gamma = dm / (1.0 - (alpha * alpha * dm));  // :368
  for (auto v : gr.nodes) g.active.insert(v);
  katz_loop(gr, gamma);
  for (int64_t i = 0; i < n_nodes; ++i) out[i] = g.centralities[g.iteration].at(nodes[i]);
}

// IsInconsistent (:322-338): both directions.
int oracle_konline_get(int64_t n_nodes, const int64_t *nodes, double *out) {
  const Map &cent = g.centralities[g.iteration];
  std::unordered_set<int64_t> ns(nodes, nodes + n_nodes);
  for (int64_t i = 0; i < n_nodes; ++i)
    if (!cent.count(nodes[i])) return 0;
  for (auto &[id, _] : cent)
    if (!ns.count(id)) return 0;
  for (int64_t i = 0; i < n_nodes; ++i) out[i] = cent.at(nodes[i]);
  return 1;
}

void oracle_konline_update(int64_t n_nodes, const int64_t *nodes, int64_t n_edges,
                           const int64_t *src, const int64_t *dst,
                           const int64_t *created_v, int64_t n_cv,
                           const int64_t *created_e, int64_t n_ce,
                           const int64_t *created_e_idx, const int64_t *deleted_v,
                           int64_t n_dv, const int64_t *deleted_e, int64_t n_de,
                           double *out) {
  Graph gr = build(n_nodes, nodes, n_edges, src, dst);
  if (n_edges == 0) {  // :390-393
    init_ctx(gr);
    for (int64_t i = 0; i < n_nodes; ++i) out[i] = g.centralities[g.iteration].at(nodes[i]);
    return;
  }
  const double dm = (double)max_degree(gr);
  const double gamma = dm / (1.0 - (g.alpha * dm));  // :396 (no alpha^2)

  for (int64_t i = 0; i < n_cv; ++i) {  // :399-404
    g.omegas[0][created_v[i]] = 1.0;
    for (int64_t it = 0; it <= g.iteration; ++it) g.centralities[it][created_v[i]] = 0.0;
  }

  std::set<int64_t> updated;  // :407-415
  for (int64_t i = 0; i < n_ce; ++i) {
    updated.insert(created_e[2 * i]);
    updated.insert(created_e[2 * i + 1]);
  }
  for (int64_t i = 0; i < n_de; ++i) {
    updated.insert(deleted_e[2 * i]);
    updated.insert(deleted_e[2 * i + 1]);
  }

  // context_new (:417-424)
  std::vector<Map> new_omegas;
  {
    Map w0;
    for (auto v : gr.nodes) w0[v] = 1.0;
    new_omegas.push_back(std::move(w0));
  }
  std::set<int64_t> new_eids(created_e_idx, created_e_idx + n_ce);
  for (int64_t i = 1; i <= g.iteration; ++i) {
    // AddIteration for context_new
    Map wi;
    for (auto v : gr.nodes) wi[v] = 0.0;
    new_omegas.push_back(std::move(wi));
    // UpdateLevel (:253-313)
    std::queue<int64_t> q;
    for (auto v : updated) q.push(v);
    for (auto &[id, val] : g.omegas[i]) new_omegas[i][id] = val;  // :266-268
    while (!q.empty()) {
      int64_t v = q.front();
      q.pop();
      if (!gr.node_set.count(v)) continue;  // :276
      auto it = gr.out.find(v);
      if (it == gr.out.end()) continue;
      for (auto [w, eid] : it->second) {
        if (!updated.count(w)) q.push(w);
        updated.insert(w);
        if (new_eids.count(eid)) continue;  // :287
        // default 0.0 when v was not in the old graph (operator[] semantics)
        double oldv = 0.0;
        auto ov = g.omegas[i - 1].find(v);
        if (ov != g.omegas[i - 1].end()) oldv = ov->second;
        new_omegas[i][w] += new_omegas[i - 1][v] - oldv;  // :289
      }
    }
    for (int64_t e = 0; e < n_ce; ++e) {  // :294-296
      new_omegas[i][created_e[2 * e + 1]] += new_omegas[i - 1][created_e[2 * e]];
    }
    for (int64_t e = 0; e < n_de; ++e) {  // :299-301
      double oldv = 0.0;
      auto ov = g.omegas[i - 1].find(deleted_e[2 * e]);
      if (ov != g.omegas[i - 1].end()) oldv = ov->second;
      new_omegas[i][deleted_e[2 * e + 1]] -= oldv;
    }
    for (auto w : updated) {  // :304-312
      if (i != 1) {
        g.centralities[i][w] =
            g.centralities[i - 1][w] + pow(g.alpha, (double)i) * new_omegas[i][w];
      } else {
        g.centralities[i][w] +=
            pow(g.alpha, (double)i) * (new_omegas[i][w] - g.omegas[i][w]);
      }
    }
  }
  for (int64_t i = 1; i <= g.iteration; ++i)  // :427-431
    for (auto &[id, val] : new_omegas[i]) g.omegas[i][id] = val;

  for (auto w : updated) {  // :434-440 (no gamma in ur here)
    g.lr[w] = g.centralities[g.iteration][w];
    g.ur[w] = g.centralities[g.iteration][w] +
              pow(g.alpha, (double)(g.iteration + 1)) * g.omegas[g.iteration][w];
  }

  double min_lr = 1e300;  // :443-448
  for (auto v : g.active) {
    double l = g.lr[v];
    if (l < min_lr) min_lr = l;
  }
  for (auto v : gr.nodes) {  // :449-455
    if (g.ur[v] >= (min_lr - g.eps)) g.active.insert(v);
  }
  for (int64_t i = 0; i < n_dv; ++i) {  // :458-464
    for (int64_t it = 0; it <= g.iteration; ++it) {
      g.omegas[it].erase(deleted_v[i]);
      g.centralities[it].erase(deleted_v[i]);
    }
    g.active.erase(deleted_v[i]);
  }

  katz_loop(gr, gamma);
  for (int64_t i = 0; i < n_nodes; ++i) out[i] = g.centralities[g.iteration].at(nodes[i]);
}

}  // extern "C"
