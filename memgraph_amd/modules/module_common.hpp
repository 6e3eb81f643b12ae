// Shared host-side plumbing for the four drop-in module .so's:
//  - thin throwing wrappers over the restated mgp C ABI (include/mgx_mgp.h),
//    mirroring the reference's _mgp.hpp convention (errors -> exceptions,
//    never across the ABI — pagerank_module.cpp:108-112 pattern);
//  - the MVCC scan -> dense COO renumbering (the reference's own approach:
//    pagerank_module.cpp:18-54 scan-order ids; louvain.cpp:74-118 first-seen
//    ids — the mode matters for Louvain's observable community numbering);
//  - result emission mirroring mg_utils.hpp:256-316 ownership (value made,
//    inserted, destroyed; vertex consumed by value_make_vertex);
//  - the process-global GPU context (fail-loud: no CPU fallback).
#pragma once

#include <cstdint>
#include <mutex>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

#include "mgx_analytics.h"
#include "mgx_mgp.h"

namespace mgx_module {

class MgpError : public std::runtime_error {
 public:
  explicit MgpError(const std::string &what) : std::runtime_error(what) {}
};

inline void Check(enum mgp_error e, const char *what) {
  if (e != MGP_ERROR_NO_ERROR) throw MgpError(std::string(what) + " failed");
}

class MgxError : public std::runtime_error {
 public:
  explicit MgxError(const std::string &what) : std::runtime_error(what) {}
};

inline void CheckMgx(mgx_status s, const char *what) {
  if (s != MGX_OK) {
    throw MgxError(std::string(what) + ": " + mgx_status_string(s) + " (" +
                   mgx_last_error() + ")");
  }
}

// Process-global GPU context, created on first procedure call. The module
// must fail loudly (via mgp_result_set_error_msg at the call site) when no
// HIP device is present — never fall back to CPU.
inline mgx_context *Ctx() {
  static mgx_context *ctx = nullptr;
  static std::mutex mu;
  std::lock_guard<std::mutex> lock(mu);
  if (!ctx) {
    const char *dev_env = getenv("MGX_DEVICE");
    const int device = dev_env ? atoi(dev_env) : 0;
    CheckMgx(mgx_init(device, &ctx), "mgx_init");
  }
  return ctx;
}

struct ScanResult {
  std::vector<int64_t> src, dst;      // dense ids
  std::vector<double> weights;        // only when read_weights
  std::vector<int64_t> dense_to_mg;   // dense id -> memgraph id
};

enum class Numbering {
  kVertexScanOrder,  // pagerank_module.cpp:41-42: dense id = scan position
  kFirstSeen,        // louvain.cpp:86-117: sources/destinations as seen
};

// The MVCC scan hot loop, through the same iterator
// ABI the reference modules use.
inline ScanResult ScanGraph(mgp_graph *graph, mgp_memory *memory, Numbering numbering,
                            bool read_weights = false, const char *weight_property = nullptr,
                            double default_weight = 1.0) {
  ScanResult out;
  size_t approx_v = 0, approx_e = 0;
  Check(mgp_graph_approximate_vertex_count(graph, &approx_v), "approx_vertex_count");
  Check(mgp_graph_approximate_edge_count(graph, &approx_e), "approx_edge_count");
  out.dense_to_mg.reserve(approx_v);
  out.src.reserve(approx_e);
  out.dst.reserve(approx_e);
  std::unordered_map<int64_t, int64_t> mg_to_dense;
  mg_to_dense.reserve(approx_v * 2);

  auto intern = [&](int64_t mg_id) -> int64_t {
    auto it = mg_to_dense.find(mg_id);
    if (it != mg_to_dense.end()) return it->second;
    const int64_t dense = (int64_t)out.dense_to_mg.size();
    mg_to_dense.emplace(mg_id, dense);
    out.dense_to_mg.push_back(mg_id);
    return dense;
  };

  struct VerticesGuard {
    mgp_vertices_iterator *it = nullptr;
    ~VerticesGuard() {
      if (it) mgp_vertices_iterator_destroy(it);
    }
  } vg;
  Check(mgp_graph_iter_vertices(graph, memory, &vg.it), "graph_iter_vertices");

  mgp_vertex *v = nullptr;
  Check(mgp_vertices_iterator_get(vg.it, &v), "vertices_iterator_get");
  while (v) {
    if (mgp_must_abort(graph)) throw MgpError("query aborted");
    mgp_vertex_id vid{0};
    Check(mgp_vertex_get_id(v, &vid), "vertex_get_id");
    int64_t source_dense = -1;
    if (numbering == Numbering::kFirstSeen) source_dense = intern(vid.as_int);

    struct EdgesGuard {
      mgp_edges_iterator *it = nullptr;
      ~EdgesGuard() {
        if (it) mgp_edges_iterator_destroy(it);
      }
    } eg;
    Check(mgp_vertex_iter_out_edges(v, memory, &eg.it), "vertex_iter_out_edges");
    mgp_edge *e = nullptr;
    Check(mgp_edges_iterator_get(eg.it, &e), "edges_iterator_get");
    while (e) {
      mgp_vertex *to = nullptr;
      Check(mgp_edge_get_to(e, &to), "edge_get_to");
      mgp_vertex_id tid{0};
      Check(mgp_vertex_get_id(to, &tid), "vertex_get_id(to)");
      if (numbering == Numbering::kFirstSeen) {
        out.src.push_back(source_dense);
        out.dst.push_back(intern(tid.as_int));
      } else {
        // Scan-order numbering: record memgraph ids now, remap after the
        // scan (the reference does exactly this — pagerank_module.cpp:39,49-52).
        out.src.push_back(vid.as_int);
        out.dst.push_back(tid.as_int);
      }
      if (read_weights) {
        // mg_utility::GetNumericProperty (mg_utils.hpp:350-368): numeric
        // property value, else default.
        double w = default_weight;
        mgp_value *pv = nullptr;
        if (mgp_edge_get_property(e, weight_property, memory, &pv) ==
                MGP_ERROR_NO_ERROR && pv) {
          int is = 0;
          int64_t iv = 0;
          double dv = 0.0;
          if (mgp_value_is_double(pv, &is) == MGP_ERROR_NO_ERROR && is &&
              mgp_value_get_double(pv, &dv) == MGP_ERROR_NO_ERROR) {
            w = dv;
          } else if (mgp_value_is_int(pv, &is) == MGP_ERROR_NO_ERROR && is &&
                     mgp_value_get_int(pv, &iv) == MGP_ERROR_NO_ERROR) {
            w = (double)iv;
          }
          mgp_value_destroy(pv);
        }
        out.weights.push_back(w);
      }
      Check(mgp_edges_iterator_next(eg.it, &e), "edges_iterator_next");
    }
    if (numbering == Numbering::kVertexScanOrder) intern(vid.as_int);
    Check(mgp_vertices_iterator_next(vg.it, &v), "vertices_iterator_next");
  }

  if (numbering == Numbering::kVertexScanOrder) {
    for (auto &s : out.src) s = mg_to_dense.at(s);
    for (auto &d : out.dst) d = mg_to_dense.at(d);
  }
  return out;
}

// Subgraph scan mirroring louvain_alg::GetLouvainSubgraph
// (louvain.cpp:120-156): nodes list interned first (first-seen dense ids,
// duplicates deduped), then edges appended PER LIST ENTRY (duplicates
// create multi-edges, exactly as the reference does) when both endpoints
// are in the node set.
inline ScanResult ScanSubgraph(mgp_graph *graph, mgp_memory *memory, mgp_list *nodes,
                               mgp_list *edges, bool read_weights,
                               const char *weight_property, double default_weight) {
  ScanResult out;
  std::unordered_map<int64_t, int64_t> mg_to_dense;
  size_t n_nodes = 0, n_edges = 0;
  Check(mgp_list_size(nodes, &n_nodes), "list_size(nodes)");
  Check(mgp_list_size(edges, &n_edges), "list_size(edges)");
  for (size_t i = 0; i < n_nodes; ++i) {
    if (mgp_must_abort(graph)) throw MgpError("query aborted");
    mgp_value *v = nullptr;
    Check(mgp_list_at(nodes, i, &v), "list_at(nodes)");
    mgp_vertex *vert = nullptr;
    Check(mgp_value_get_vertex(v, &vert), "value_get_vertex");
    mgp_vertex_id vid{0};
    Check(mgp_vertex_get_id(vert, &vid), "vertex_get_id");
    if (!mg_to_dense.count(vid.as_int)) {
      mg_to_dense.emplace(vid.as_int, (int64_t)out.dense_to_mg.size());
      out.dense_to_mg.push_back(vid.as_int);
    }
  }
  for (size_t i = 0; i < n_edges; ++i) {
    if (mgp_must_abort(graph)) throw MgpError("query aborted");
    mgp_value *v = nullptr;
    Check(mgp_list_at(edges, i, &v), "list_at(edges)");
    mgp_edge *e = nullptr;
    Check(mgp_value_get_edge(v, &e), "value_get_edge");
    mgp_vertex *from = nullptr, *to = nullptr;
    Check(mgp_edge_get_from(e, &from), "edge_get_from");
    Check(mgp_edge_get_to(e, &to), "edge_get_to");
    mgp_vertex_id fid{0}, tid{0};
    Check(mgp_vertex_get_id(from, &fid), "vertex_get_id(from)");
    Check(mgp_vertex_get_id(to, &tid), "vertex_get_id(to)");
    auto fit = mg_to_dense.find(fid.as_int);
    auto tit = mg_to_dense.find(tid.as_int);
    if (fit == mg_to_dense.end() || tit == mg_to_dense.end()) continue;
    out.src.push_back(fit->second);
    out.dst.push_back(tit->second);
    if (read_weights) {
      double w = default_weight;
      mgp_value *pv = nullptr;
      if (mgp_edge_get_property(e, weight_property, memory, &pv) == MGP_ERROR_NO_ERROR &&
          pv) {
        int is = 0;
        int64_t iv = 0;
        double dv = 0.0;
        if (mgp_value_is_double(pv, &is) == MGP_ERROR_NO_ERROR && is &&
            mgp_value_get_double(pv, &dv) == MGP_ERROR_NO_ERROR) {
          w = dv;
        } else if (mgp_value_is_int(pv, &is) == MGP_ERROR_NO_ERROR && is &&
                   mgp_value_get_int(pv, &iv) == MGP_ERROR_NO_ERROR) {
          w = (double)iv;
        }
        mgp_value_destroy(pv);
      }
      out.weights.push_back(w);
    }
  }
  return out;
}

// Emission mirroring InsertPagerankRecord / InsertWeaklyComponentResult
// ownership (mg_utils.hpp:256-316): nullptr vertex in non-transactional
// storage is skipped, in transactional storage it is an error.
template <typename InsertValue>
inline void EmitNodeRecord(mgp_graph *graph, mgp_result *result, mgp_memory *memory,
                           int64_t mg_id, const char *node_field, InsertValue &&insert_value) {
  mgp_vertex *vertex = nullptr;
  enum mgp_error ge = mgp_graph_get_vertex_by_id(graph, mgp_vertex_id{mg_id}, memory,
                                                 &vertex);
  if (ge != MGP_ERROR_NO_ERROR || !vertex) {
    int transactional = 1;
    Check(mgp_graph_is_transactional(graph, &transactional), "graph_is_transactional");
    if (transactional) throw MgpError("invalid vertex id during result emission");
    return;
  }
  mgp_result_record *record = nullptr;
  Check(mgp_result_new_record(result, &record), "result_new_record");
  mgp_value *vval = nullptr;
  Check(mgp_value_make_vertex(vertex, &vval), "value_make_vertex");
  Check(mgp_result_record_insert(record, node_field, vval), "record_insert(node)");
  mgp_value_destroy(vval);
  insert_value(record);
}

inline void InsertDouble(mgp_result_record *record, const char *field, double v,
                         mgp_memory *memory) {
  mgp_value *val = nullptr;
  Check(mgp_value_make_double(v, memory, &val), "value_make_double");
  Check(mgp_result_record_insert(record, field, val), "record_insert(double)");
  mgp_value_destroy(val);
}

inline void InsertInt(mgp_result_record *record, const char *field, int64_t v,
                      mgp_memory *memory) {
  mgp_value *val = nullptr;
  Check(mgp_value_make_int(v, memory, &val), "value_make_int");
  Check(mgp_result_record_insert(record, field, val), "record_insert(int)");
  mgp_value_destroy(val);
}

inline void InsertString(mgp_result_record *record, const char *field, const char *v,
                         mgp_memory *memory) {
  mgp_value *val = nullptr;
  Check(mgp_value_make_string(v, memory, &val), "value_make_string");
  Check(mgp_result_record_insert(record, field, val), "record_insert(string)");
  mgp_value_destroy(val);
}

inline void InsertIntList(mgp_result_record *record, const char *field,
                          const std::vector<int64_t> &vals, mgp_memory *memory) {
  mgp_list *list = nullptr;
  Check(mgp_list_make_empty(vals.size(), memory, &list), "list_make_empty");
  for (int64_t v : vals) {
    mgp_value *iv = nullptr;
    Check(mgp_value_make_int(v, memory, &iv), "value_make_int");
    Check(mgp_list_append_extend(list, iv), "list_append_extend");
    mgp_value_destroy(iv);
  }
  mgp_value *lv = nullptr;
  Check(mgp_value_make_list(list, &lv), "value_make_list");
  Check(mgp_result_record_insert(record, field, lv), "record_insert(list)");
  mgp_value_destroy(lv);
}

// Argument readers (procedure-supplied args arrive positionally).
inline int64_t ArgInt(mgp_list *args, size_t i) {
  mgp_value *v = nullptr;
  Check(mgp_list_at(args, i, &v), "list_at");
  int64_t out = 0;
  Check(mgp_value_get_int(v, &out), "value_get_int");
  return out;
}

inline double ArgDouble(mgp_list *args, size_t i) {
  mgp_value *v = nullptr;
  Check(mgp_list_at(args, i, &v), "list_at");
  double out = 0;
  Check(mgp_value_get_double(v, &out), "value_get_double");
  return out;
}

inline bool ArgBool(mgp_list *args, size_t i) {
  mgp_value *v = nullptr;
  Check(mgp_list_at(args, i, &v), "list_at");
  int out = 0;
  Check(mgp_value_get_bool(v, &out), "value_get_bool");
  return out != 0;
}

inline const char *ArgString(mgp_list *args, size_t i) {
  mgp_value *v = nullptr;
  Check(mgp_list_at(args, i, &v), "list_at");
  const char *out = nullptr;
  Check(mgp_value_get_string(v, &out), "value_get_string");
  return out;
}

// RAII graph handle.
struct GraphGuard {
  mgx_context *ctx;
  mgx_graph *g = nullptr;
  ~GraphGuard() {
    if (g) mgx_graph_destroy(ctx, g);
  }
};

}  // namespace mgx_module
