// TEST INFRASTRUCTURE ONLY — host-mock of the mgp C ABI.
//
// Implements the subset of mg_procedure.h that the four drop-in module .so's
// import (declared in include/mgx_mgp.h), over an in-memory edge list —
// the pattern of the reference's own ABI tests
// (tests/unit/query_procedures_mgp_graph.cpp:123-146, which build an
// mgp_graph straight from a storage accessor). Lets pytest dlopen the REAL
// module .so's (RTLD_GLOBAL mock first, then the module resolves mgp_*
// here) and drive CALL-like invocations without a memgraphd.
//
// Also exports a mock_* driver API for ctypes: build graph, invoke a
// registered procedure with its default (or overridden) arguments, read
// back result rows.

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "mgx_mgp.h"

namespace {

struct Value;

struct MockEdgeProp {
  std::string name;
  double value;
};

struct MockEdge {
  int64_t from_mg, to_mg;
  bool has_prop = false;
  MockEdgeProp prop;
};

struct MockVertexRec {
  int64_t mg_id;
  std::vector<int64_t> out_edges;  // indices into edges
};

struct MockGraph {
  std::vector<MockVertexRec> vertices;  // scan order
  std::map<int64_t, int64_t> mg_to_idx;
  std::vector<MockEdge> edges;
};

MockGraph g_graph;

struct ResultRow {
  std::map<std::string, Value> fields;
};

struct MockResult {
  std::vector<ResultRow> rows;
  std::string error;
  bool has_error = false;
};

MockResult g_result;

}  // namespace

// ---- opaque ABI types --------------------------------------------------

struct mgp_vertex {
  int64_t mg_id;
};

struct mgp_edge {
  const MockEdge *edge;
  mgp_vertex from_v, to_v;
};

namespace {
struct Value {
  enum Kind { kNull, kBool, kInt, kDouble, kString, kVertex, kEdge, kList } kind = kNull;
  int64_t i = 0;
  double d = 0.0;
  std::string s;
  mgp_vertex *vertex = nullptr;  // owned when kind == kVertex
  struct mgp_edge *edge = nullptr;   // arena-owned
  struct mgp_list *list = nullptr;   // arena-owned
};
}  // namespace

struct mgp_value {
  Value v;
};

struct mgp_list {
  std::vector<mgp_value *> items;
};

struct mgp_memory {
  int dummy;
};

struct mgp_graph {
  MockGraph *g;
};

struct mgp_result {
  MockResult *r;
};

struct mgp_result_record {
  ResultRow *row;
};

struct mgp_vertices_iterator {
  MockGraph *g;
  size_t pos = 0;
  mgp_vertex cur;
};

struct mgp_edges_iterator {
  MockGraph *g;
  const MockVertexRec *v;
  size_t pos = 0;
  mgp_edge cur;
};

struct mgp_type {
  const char *name;
};

struct ProcArg {
  std::string name;
  mgp_type *type;
  bool optional = false;
  Value default_value;
};

struct mgp_proc {
  std::string name;
  mgp_proc_cb cb;
  std::vector<ProcArg> args;
  std::vector<std::pair<std::string, mgp_type *>> results;
};

struct mgp_module {
  std::vector<std::unique_ptr<mgp_proc>> procs;
};

namespace {
mgp_module g_module;
mgp_memory g_memory;
// Arena for list/edge/value objects handed to procedures via arg overrides;
// cleared on mock_reset*/mock_call completion boundaries.
std::vector<std::unique_ptr<mgp_list>> g_arena_lists;
std::vector<std::unique_ptr<mgp_value>> g_arena_values;
std::vector<std::unique_ptr<mgp_edge>> g_arena_edges;
std::vector<std::unique_ptr<mgp_vertex>> g_arena_vertices;

void clear_arena() {
  g_arena_lists.clear();
  g_arena_values.clear();
  g_arena_edges.clear();
  g_arena_vertices.clear();
}
mgp_graph g_graph_handle{&g_graph};
mgp_result g_result_handle{&g_result};
std::vector<std::pair<size_t, Value>> g_arg_overrides;

mgp_type g_t_bool{"bool"}, g_t_string{"string"}, g_t_int{"int"}, g_t_float{"float"},
    g_t_node{"node"}, g_t_relationship{"relationship"}, g_t_list{"list"},
    g_t_nullable{"nullable"};
// Registration-time lists (empty-list defaults for the online update() args)
// live for the process: ProcArg::default_value copies the raw pointer.
std::vector<std::unique_ptr<mgp_list>> g_perm_lists;
}  // namespace

// ---- ABI implementation ------------------------------------------------

extern "C" {

enum mgp_error mgp_value_make_bool(int val, struct mgp_memory *, struct mgp_value **result) {
  auto *v = new mgp_value();
  v->v.kind = Value::kBool;
  v->v.i = val ? 1 : 0;
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_make_int(int64_t val, struct mgp_memory *, struct mgp_value **result) {
  auto *v = new mgp_value();
  v->v.kind = Value::kInt;
  v->v.i = val;
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_make_double(double val, struct mgp_memory *,
                                     struct mgp_value **result) {
  auto *v = new mgp_value();
  v->v.kind = Value::kDouble;
  v->v.d = val;
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_make_string(const char *val, struct mgp_memory *,
                                     struct mgp_value **result) {
  auto *v = new mgp_value();
  v->v.kind = Value::kString;
  v->v.s = val;
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_make_vertex(struct mgp_vertex *val, struct mgp_value **result) {
  auto *v = new mgp_value();
  v->v.kind = Value::kVertex;
  v->v.vertex = val;  // takes ownership (mg_procedure.h:270-272)
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

void mgp_value_destroy(struct mgp_value *val) {
  if (!val) return;
  if (val->v.kind == Value::kVertex && val->v.vertex) delete val->v.vertex;
  delete val;
}

enum mgp_error mgp_value_is_null(struct mgp_value *val, int *result) {
  *result = val->v.kind == Value::kNull;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_is_int(struct mgp_value *val, int *result) {
  *result = val->v.kind == Value::kInt;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_is_double(struct mgp_value *val, int *result) {
  *result = val->v.kind == Value::kDouble;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_bool(struct mgp_value *val, int *result) {
  if (val->v.kind != Value::kBool) return MGP_ERROR_LOGIC_ERROR;
  *result = (int)val->v.i;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_int(struct mgp_value *val, int64_t *result) {
  if (val->v.kind != Value::kInt) return MGP_ERROR_LOGIC_ERROR;
  *result = val->v.i;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_double(struct mgp_value *val, double *result) {
  if (val->v.kind != Value::kDouble) return MGP_ERROR_LOGIC_ERROR;
  *result = val->v.d;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_string(struct mgp_value *val, const char **result) {
  if (val->v.kind != Value::kString) return MGP_ERROR_LOGIC_ERROR;
  *result = val->v.s.c_str();
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_list(struct mgp_value *val, struct mgp_list **result) {
  if (val->v.kind != Value::kList) return MGP_ERROR_LOGIC_ERROR;
  *result = val->v.list;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_vertex(struct mgp_value *val, struct mgp_vertex **result) {
  if (val->v.kind != Value::kVertex) return MGP_ERROR_LOGIC_ERROR;
  *result = val->v.vertex;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_get_edge(struct mgp_value *val, struct mgp_edge **result) {
  if (val->v.kind != Value::kEdge) return MGP_ERROR_LOGIC_ERROR;
  *result = val->v.edge;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_list_make_empty(size_t capacity, struct mgp_memory *,
                                   struct mgp_list **result) {
  (void)capacity;
  g_perm_lists.push_back(std::make_unique<mgp_list>());
  *result = g_perm_lists.back().get();
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_value_make_list(struct mgp_list *val, struct mgp_value **result) {
  auto *v = new mgp_value();
  v->v.kind = Value::kList;
  v->v.list = val;
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

// Enterprise gate used by the online modules (mg_procedure.h:65). The mock
// host is "licensed" unless MOCK_ENTERPRISE=0 (for gate tests).
int mgp_is_enterprise_valid(void) {
  const char *e = getenv("MOCK_ENTERPRISE");
  return (e && e[0] == '0') ? 0 : 1;
}

enum mgp_error mgp_list_append_extend(struct mgp_list *list, struct mgp_value *val) {
  // mg_procedure.h:544: append, extending capacity as needed; the mock
  // copies the value into the process arena so result lists stay valid.
  auto v = std::make_unique<mgp_value>();
  v->v = val->v;
  list->items.push_back(v.get());
  g_arena_values.push_back(std::move(v));
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_list_size(struct mgp_list *list, size_t *result) {
  *result = list->items.size();
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_list_at(struct mgp_list *list, size_t index, struct mgp_value **result) {
  if (index >= list->items.size()) return MGP_ERROR_OUT_OF_RANGE;
  *result = list->items[index];
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_result_set_error_msg(struct mgp_result *res, const char *error_msg) {
  res->r->has_error = true;
  res->r->error = error_msg ? error_msg : "";
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_result_new_record(struct mgp_result *res, struct mgp_result_record **result) {
  res->r->rows.emplace_back();
  auto *rec = new mgp_result_record();
  rec->row = &res->r->rows.back();
  *result = rec;  // leaked per-call in mock; reset by mock_reset
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_result_record_insert(struct mgp_result_record *record,
                                        const char *field_name, struct mgp_value *val) {
  Value copy = val->v;
  if (copy.kind == Value::kVertex) {
    // copy the vertex payload; ownership of the original stays with val
    copy.i = val->v.vertex->mg_id;
    copy.vertex = nullptr;
  }
  record->row->fields[field_name] = copy;
  return MGP_ERROR_NO_ERROR;
}

void mgp_vertex_destroy(struct mgp_vertex *v) { delete v; }

enum mgp_error mgp_vertex_get_id(struct mgp_vertex *v, struct mgp_vertex_id *result) {
  result->as_int = v->mg_id;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_vertex_iter_out_edges(struct mgp_vertex *v, struct mgp_memory *,
                                         struct mgp_edges_iterator **result) {
  auto *it = new mgp_edges_iterator();
  it->g = &g_graph;
  auto idx_it = g_graph.mg_to_idx.find(v->mg_id);
  if (idx_it == g_graph.mg_to_idx.end()) return MGP_ERROR_INVALID_ARGUMENT;
  it->v = &g_graph.vertices[idx_it->second];
  it->pos = 0;
  *result = it;
  return MGP_ERROR_NO_ERROR;
}

void mgp_edges_iterator_destroy(struct mgp_edges_iterator *it) { delete it; }

static void fill_edge(mgp_edges_iterator *it) {
  const MockEdge &e = it->g->edges[it->v->out_edges[it->pos]];
  it->cur.edge = &e;
  it->cur.from_v.mg_id = e.from_mg;
  it->cur.to_v.mg_id = e.to_mg;
}

enum mgp_error mgp_edges_iterator_get(struct mgp_edges_iterator *it,
                                      struct mgp_edge **result) {
  if (it->pos >= it->v->out_edges.size()) {
    *result = nullptr;
    return MGP_ERROR_NO_ERROR;
  }
  fill_edge(it);
  *result = &it->cur;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_edges_iterator_next(struct mgp_edges_iterator *it,
                                       struct mgp_edge **result) {
  ++it->pos;
  return mgp_edges_iterator_get(it, result);
}

enum mgp_error mgp_edge_get_from(struct mgp_edge *e, struct mgp_vertex **result) {
  *result = &e->from_v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_edge_get_to(struct mgp_edge *e, struct mgp_vertex **result) {
  *result = &e->to_v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_edge_get_property(struct mgp_edge *e, const char *property_name,
                                     struct mgp_memory *, struct mgp_value **result) {
  auto *v = new mgp_value();
  if (e->edge->has_prop && e->edge->prop.name == property_name) {
    v->v.kind = Value::kDouble;
    v->v.d = e->edge->prop.value;
  } else {
    v->v.kind = Value::kNull;
  }
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_graph_get_vertex_by_id(struct mgp_graph *g, struct mgp_vertex_id id,
                                          struct mgp_memory *, struct mgp_vertex **result) {
  auto it = g->g->mg_to_idx.find(id.as_int);
  if (it == g->g->mg_to_idx.end()) {
    *result = nullptr;
    return MGP_ERROR_NO_ERROR;
  }
  auto *v = new mgp_vertex();
  v->mg_id = id.as_int;
  *result = v;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_graph_is_transactional(struct mgp_graph *, int *result) {
  *result = 1;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_graph_iter_vertices(struct mgp_graph *g, struct mgp_memory *,
                                       struct mgp_vertices_iterator **result) {
  auto *it = new mgp_vertices_iterator();
  it->g = g->g;
  it->pos = 0;
  *result = it;
  return MGP_ERROR_NO_ERROR;
}

void mgp_vertices_iterator_destroy(struct mgp_vertices_iterator *it) { delete it; }

enum mgp_error mgp_vertices_iterator_get(struct mgp_vertices_iterator *it,
                                         struct mgp_vertex **result) {
  if (it->pos >= it->g->vertices.size()) {
    *result = nullptr;
    return MGP_ERROR_NO_ERROR;
  }
  it->cur.mg_id = it->g->vertices[it->pos].mg_id;
  *result = &it->cur;
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_vertices_iterator_next(struct mgp_vertices_iterator *it,
                                          struct mgp_vertex **result) {
  ++it->pos;
  return mgp_vertices_iterator_get(it, result);
}

enum mgp_error mgp_graph_approximate_vertex_count(struct mgp_graph *g, size_t *result) {
  *result = g->g->vertices.size();
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_graph_approximate_edge_count(struct mgp_graph *g, size_t *result) {
  *result = g->g->edges.size();
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_type_bool(struct mgp_type **result) { *result = &g_t_bool; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_string(struct mgp_type **result) { *result = &g_t_string; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_int(struct mgp_type **result) { *result = &g_t_int; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_float(struct mgp_type **result) { *result = &g_t_float; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_node(struct mgp_type **result) { *result = &g_t_node; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_relationship(struct mgp_type **result) { *result = &g_t_relationship; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_list(struct mgp_type *, struct mgp_type **result) { *result = &g_t_list; return MGP_ERROR_NO_ERROR; }
enum mgp_error mgp_type_nullable(struct mgp_type *, struct mgp_type **result) { *result = &g_t_nullable; return MGP_ERROR_NO_ERROR; }

enum mgp_error mgp_module_add_read_procedure(struct mgp_module *module, const char *name,
                                             mgp_proc_cb cb, struct mgp_proc **result) {
  auto proc = std::make_unique<mgp_proc>();
  proc->name = name;
  proc->cb = cb;
  *result = proc.get();
  module->procs.push_back(std::move(proc));
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_proc_add_arg(struct mgp_proc *proc, const char *name,
                                struct mgp_type *type) {
  proc->args.push_back({name, type, false, Value{}});
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_proc_add_opt_arg(struct mgp_proc *proc, const char *name,
                                    struct mgp_type *type, struct mgp_value *default_value) {
  proc->args.push_back({name, type, true, default_value->v});
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_proc_add_result(struct mgp_proc *proc, const char *name,
                                   struct mgp_type *type) {
  proc->results.emplace_back(name, type);
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_track_current_thread_allocations(struct mgp_graph *) {
  return MGP_ERROR_NO_ERROR;
}

enum mgp_error mgp_untrack_current_thread_allocations(struct mgp_graph *) {
  return MGP_ERROR_NO_ERROR;
}

int mgp_must_abort(struct mgp_graph *) { return 0; }

// ---- mock driver API (ctypes) ------------------------------------------

void mock_reset(void) {
  g_graph = MockGraph{};
  g_result = MockResult{};
  g_module.procs.clear();
  g_arg_overrides.clear();
  clear_arena();
}

void mock_reset_graph(void) {
  g_graph = MockGraph{};
  g_result = MockResult{};
  g_arg_overrides.clear();
  clear_arena();
}

void mock_add_vertex(int64_t mg_id) {
  if (g_graph.mg_to_idx.count(mg_id)) return;
  g_graph.mg_to_idx[mg_id] = (int64_t)g_graph.vertices.size();
  g_graph.vertices.push_back({mg_id, {}});
}

void mock_add_edge(int64_t from_mg, int64_t to_mg) {
  mock_add_vertex(from_mg);
  mock_add_vertex(to_mg);
  MockEdge e{from_mg, to_mg, false, {}};
  g_graph.vertices[g_graph.mg_to_idx[from_mg]].out_edges.push_back(
      (int64_t)g_graph.edges.size());
  g_graph.edges.push_back(e);
}

void mock_add_edge_weighted(int64_t from_mg, int64_t to_mg, const char *prop_name,
                            double weight) {
  mock_add_vertex(from_mg);
  mock_add_vertex(to_mg);
  MockEdge e{from_mg, to_mg, true, {prop_name, weight}};
  g_graph.vertices[g_graph.mg_to_idx[from_mg]].out_edges.push_back(
      (int64_t)g_graph.edges.size());
  g_graph.edges.push_back(e);
}

void *mock_module(void) { return &g_module; }
void *mock_memory(void) { return &g_memory; }

int64_t mock_proc_count(void) { return (int64_t)g_module.procs.size(); }

const char *mock_proc_name(int64_t i) { return g_module.procs[i]->name.c_str(); }

int64_t mock_proc_arg_count(const char *proc_name) {
  for (auto &p : g_module.procs)
    if (p->name == proc_name) return (int64_t)p->args.size();
  return -1;
}

const char *mock_proc_arg_name(const char *proc_name, int64_t i) {
  for (auto &p : g_module.procs)
    if (p->name == proc_name) return p->args[i].name.c_str();
  return nullptr;
}

const char *mock_proc_arg_type(const char *proc_name, int64_t i) {
  for (auto &p : g_module.procs)
    if (p->name == proc_name) return p->args[i].type->name;
  return nullptr;
}

int64_t mock_proc_result_count(const char *proc_name) {
  for (auto &p : g_module.procs)
    if (p->name == proc_name) return (int64_t)p->results.size();
  return -1;
}

const char *mock_proc_result_name(const char *proc_name, int64_t i) {
  for (auto &p : g_module.procs)
    if (p->name == proc_name) return p->results[i].first.c_str();
  return nullptr;
}

const char *mock_proc_result_type(const char *proc_name, int64_t i) {
  for (auto &p : g_module.procs)
    if (p->name == proc_name) return p->results[i].second->name;
  return nullptr;
}

void mock_override_arg_bool(int64_t pos, int64_t v) {
  Value val;
  val.kind = Value::kBool;
  val.i = v ? 1 : 0;
  g_arg_overrides.emplace_back((size_t)pos, val);
}

void mock_override_arg_int(int64_t pos, int64_t v) {
  Value val;
  val.kind = Value::kInt;
  val.i = v;
  g_arg_overrides.emplace_back((size_t)pos, val);
}

void mock_override_arg_double(int64_t pos, double v) {
  Value val;
  val.kind = Value::kDouble;
  val.d = v;
  g_arg_overrides.emplace_back((size_t)pos, val);
}

// Override an argument with a list of node values (by memgraph id).
void mock_override_arg_node_list(int64_t pos, const int64_t *ids, int64_t n) {
  auto list = std::make_unique<mgp_list>();
  for (int64_t i = 0; i < n; ++i) {
    auto vert = std::make_unique<mgp_vertex>();
    vert->mg_id = ids[i];
    auto val = std::make_unique<mgp_value>();
    val->v.kind = Value::kVertex;
    val->v.vertex = vert.get();
    list->items.push_back(val.get());
    g_arena_vertices.push_back(std::move(vert));
    g_arena_values.push_back(std::move(val));
  }
  Value lv;
  lv.kind = Value::kList;
  lv.list = list.get();
  g_arena_lists.push_back(std::move(list));
  g_arg_overrides.emplace_back((size_t)pos, lv);
}

// Override an argument with a list of relationship values given as
// (from, to) memgraph-id pairs; properties come from the first matching
// graph edge (if any).
void mock_override_arg_edge_list(int64_t pos, const int64_t *from_ids,
                                 const int64_t *to_ids, int64_t n) {
  auto list = std::make_unique<mgp_list>();
  for (int64_t i = 0; i < n; ++i) {
    auto edge = std::make_unique<mgp_edge>();
    static MockEdge fallback;
    const MockEdge *src_edge = &fallback;
    for (const auto &e : g_graph.edges) {
      if (e.from_mg == from_ids[i] && e.to_mg == to_ids[i]) {
        src_edge = &e;
        break;
      }
    }
    edge->edge = src_edge;
    edge->from_v.mg_id = from_ids[i];
    edge->to_v.mg_id = to_ids[i];
    auto val = std::make_unique<mgp_value>();
    val->v.kind = Value::kEdge;
    val->v.edge = edge.get();
    list->items.push_back(val.get());
    g_arena_edges.push_back(std::move(edge));
    g_arena_values.push_back(std::move(val));
  }
  Value lv;
  lv.kind = Value::kList;
  lv.list = list.get();
  g_arena_lists.push_back(std::move(list));
  g_arg_overrides.emplace_back((size_t)pos, lv);
}

void mock_override_arg_string(int64_t pos, const char *v) {
  Value val;
  val.kind = Value::kString;
  val.s = v;
  g_arg_overrides.emplace_back((size_t)pos, val);
}

// Invoke a registered procedure with default (or overridden) args.
// Returns 0 on success, 1 if the procedure set an error, -1 if not found.
int mock_call(const char *proc_name) {
  g_result = MockResult{};
  for (auto &p : g_module.procs) {
    if (p->name != proc_name) continue;
    mgp_list args;
    std::vector<std::unique_ptr<mgp_value>> storage;
    for (auto &a : p->args) {
      auto v = std::make_unique<mgp_value>();
      v->v = a.default_value;
      storage.push_back(std::move(v));
      args.items.push_back(storage.back().get());
    }
    for (auto &[pos, val] : g_arg_overrides) {
      if (pos < args.items.size()) args.items[pos]->v = val;
    }
    g_arg_overrides.clear();
    p->cb(&args, &g_graph_handle, &g_result_handle, &g_memory);
    return g_result.has_error ? 1 : 0;
  }
  return -1;
}

const char *mock_result_error(void) { return g_result.error.c_str(); }

int64_t mock_result_count(void) { return (int64_t)g_result.rows.size(); }

// For node fields the stored payload is the memgraph id.
int64_t mock_result_int(int64_t row, const char *field) {
  auto &f = g_result.rows[row].fields.at(field);
  return f.i;
}

double mock_result_double(int64_t row, const char *field) {
  auto &f = g_result.rows[row].fields.at(field);
  return f.d;
}

const char *mock_result_string(int64_t row, const char *field) {
  auto &f = g_result.rows[row].fields.at(field);
  return f.s.c_str();
}

int64_t mock_result_list_len(int64_t row, const char *field) {
  auto &f = g_result.rows[row].fields.at(field);
  if (!f.list) return -1;
  return (int64_t)f.list->items.size();
}

int64_t mock_result_list_int(int64_t row, const char *field, int64_t i) {
  auto &f = g_result.rows[row].fields.at(field);
  return f.list->items[i]->v.i;
}

}  // extern "C"
