// leiden_community_detection.so — drop-in replacement for the reference
// leiden module (src/mage/cpp/leiden_community_detection_module/
// leiden_community_detection_module.cpp), GPU-backed
// (memgraph_amd/csrc/leiden.hip).
//
// Procedures reproduced exactly (:108-160):
//   leiden_community_detection.get(weight_property="weight":string,
//       gamma=1.0:float, theta=0.01:float, resolution_parameter=0.01:float,
//       number_of_iterations=<u64max as int>:int)
//       -> (node: node, community_id: int, communities: list<int>)
//   leiden_community_detection.get_subgraph(subgraph_nodes,
//       subgraph_relationships, <same optionals>) -> same results
// community_id = the TOP level of the node's dendrogram hierarchy;
// communities = the full bottom-up hierarchy (:39-56 InsertLeidenRecord).
// A graph whose Leiden hierarchy is empty for some node raises the
// reference's "No communities detected." error (leiden.cpp:585-586).
//
// The reference algorithm is random_device-seeded — parity is the
// DESIGN.md statistical bar. MGX_LEIDEN_SEED pins the seed for tests.

#include <random>

#include "module_common.hpp"

namespace {

using namespace mgx_module;

constexpr const char *kFieldNode = "node";
constexpr const char *kFieldCommunity = "community_id";
constexpr const char *kFieldCommunities = "communities";
constexpr int64_t kHierCap = 64;

uint64_t PickSeed() {
  const char *env = getenv("MGX_LEIDEN_SEED");
  if (env) return (uint64_t)strtoull(env, nullptr, 10);
  std::random_device rd;  // reference: std::random_device{} (leiden.cpp:65)
  return ((uint64_t)rd() << 32) ^ rd();
}

void RunLeiden(mgp_graph *graph, mgp_result *result, mgp_memory *memory,
               const ScanResult &scan, const char *weight_property, double gamma,
               double theta, double resolution, int64_t max_iterations) {
  (void)weight_property;
  const int64_t V = (int64_t)scan.dense_to_mg.size();
  // Leiden returns {} when nodes or edges are empty (leiden.cpp:485-487) —
  // the module then emits no rows.
  if (V == 0 || scan.src.empty()) return;
  if (max_iterations < 0) max_iterations = INT64_MAX;  // u64max default cast

  mgx_context *ctx = Ctx();
  GraphGuard gg{ctx};
  CheckMgx(mgx_graph_from_coo(ctx, scan.src.data(), scan.dst.data(),
                              scan.weights.empty() ? nullptr : scan.weights.data(), V,
                              (int64_t)scan.src.size(),
                              MGX_BUILD_SYM_CSR |
                                  (scan.weights.empty() ? 0u : MGX_BUILD_WEIGHTED),
                              &gg.g),
           "mgx_graph_from_coo");
  std::vector<int64_t> hier(V * kHierCap);
  std::vector<int64_t> levels(V);
  CheckMgx(mgx_leiden(ctx, gg.g, gamma, theta, resolution, max_iterations, PickSeed(),
                      kHierCap, hier.data(), levels.data()),
           "mgx_leiden");

  for (int64_t v = 0; v < V; ++v) {
    // GetCommunities throws when a node has an empty hierarchy
    // (leiden.cpp:585-586)
    if (levels[v] == 0) throw std::runtime_error("No communities detected.");
  }
  for (int64_t v = 0; v < V; ++v) {
    std::vector<int64_t> comms(hier.begin() + v * kHierCap,
                               hier.begin() + v * kHierCap + levels[v]);
    EmitNodeRecord(graph, result, memory, scan.dense_to_mg[v], kFieldNode,
                   [&](mgp_result_record *rec) {
                     InsertInt(rec, kFieldCommunity, comms.back(), memory);
                     InsertIntList(rec, kFieldCommunities, comms, memory);
                   });
  }
}

void OnGraph(mgp_list *args, mgp_graph *graph, mgp_result *result, mgp_memory *memory) {
  try {
    const char *weight_property = ArgString(args, 0);
    const double gamma = ArgDouble(args, 1);
    const double theta = ArgDouble(args, 2);
    const double resolution = ArgDouble(args, 3);
    const int64_t max_iterations = ArgInt(args, 4);
    // weighted iff any edge carries the property — mirror GetGraphView's
    // weighted variant by always reading weights with default 1.0
    ScanResult scan = ScanGraph(graph, memory, Numbering::kVertexScanOrder,
                                /*read_weights=*/true, weight_property, 1.0);
    RunLeiden(graph, result, memory, scan, weight_property, gamma, theta, resolution,
              max_iterations);
  } catch (const std::exception &e) {
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

mgp_list *ArgList(mgp_list *args, size_t i) {
  mgp_value *v = nullptr;
  Check(mgp_list_at(args, i, &v), "list_at");
  mgp_list *out = nullptr;
  Check(mgp_value_get_list(v, &out), "value_get_list");
  return out;
}

void OnSubgraph(mgp_list *args, mgp_graph *graph, mgp_result *result, mgp_memory *memory) {
  try {
    mgp_list *nodes = ArgList(args, 0);
    mgp_list *relationships = ArgList(args, 1);
    const char *weight_property = ArgString(args, 2);
    const double gamma = ArgDouble(args, 3);
    const double theta = ArgDouble(args, 4);
    const double resolution = ArgDouble(args, 5);
    const int64_t max_iterations = ArgInt(args, 6);
    ScanResult scan = ScanSubgraph(graph, memory, nodes, relationships,
                                   /*read_weights=*/true, weight_property, 1.0);
    RunLeiden(graph, result, memory, scan, weight_property, gamma, theta, resolution,
              max_iterations);
  } catch (const std::exception &e) {
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

}  // namespace

extern "C" int mgp_init_module(struct mgp_module *module, struct mgp_memory *memory) {
  mgp_value *d_wp = nullptr, *d_gamma = nullptr, *d_theta = nullptr, *d_res = nullptr,
            *d_iters = nullptr;
  try {
    mgp_type *t_int = nullptr, *t_float = nullptr, *t_string = nullptr, *t_node = nullptr,
             *t_rel = nullptr, *t_list_int = nullptr;
    Check(mgp_type_int(&t_int), "type_int");
    Check(mgp_type_float(&t_float), "type_float");
    Check(mgp_type_string(&t_string), "type_string");
    Check(mgp_type_node(&t_node), "type_node");
    Check(mgp_type_relationship(&t_rel), "type_relationship");
    Check(mgp_type_list(t_int, &t_list_int), "type_list(int)");

    Check(mgp_value_make_string("weight", memory, &d_wp), "mk");
    Check(mgp_value_make_double(1.0, memory, &d_gamma), "mk");
    Check(mgp_value_make_double(0.01, memory, &d_theta), "mk");
    Check(mgp_value_make_double(0.01, memory, &d_res), "mk");
    // kDefaultMaxIterations = uint64max cast to int64 (:36,121)
    Check(mgp_value_make_int((int64_t)UINT64_MAX, memory, &d_iters), "mk");

    mgp_proc *proc = nullptr;
    Check(mgp_module_add_read_procedure(module, "get", OnGraph, &proc), "add(get)");
    Check(mgp_proc_add_opt_arg(proc, "weight_property", t_string, d_wp), "arg");
    Check(mgp_proc_add_opt_arg(proc, "gamma", t_float, d_gamma), "arg");
    Check(mgp_proc_add_opt_arg(proc, "theta", t_float, d_theta), "arg");
    Check(mgp_proc_add_opt_arg(proc, "resolution_parameter", t_float, d_res), "arg");
    Check(mgp_proc_add_opt_arg(proc, "number_of_iterations", t_int, d_iters), "arg");
    Check(mgp_proc_add_result(proc, kFieldNode, t_node), "res");
    Check(mgp_proc_add_result(proc, kFieldCommunity, t_int), "res");
    Check(mgp_proc_add_result(proc, kFieldCommunities, t_list_int), "res");

    mgp_proc *sproc = nullptr;
    Check(mgp_module_add_read_procedure(module, "get_subgraph", OnSubgraph, &sproc),
          "add(get_subgraph)");
    mgp_type *t_list_node = nullptr, *t_list_rel = nullptr;
    Check(mgp_type_list(t_node, &t_list_node), "type_list(node)");
    Check(mgp_type_list(t_rel, &t_list_rel), "type_list(rel)");
    Check(mgp_proc_add_arg(sproc, "subgraph_nodes", t_list_node), "arg");
    Check(mgp_proc_add_arg(sproc, "subgraph_relationships", t_list_rel), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "weight_property", t_string, d_wp), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "gamma", t_float, d_gamma), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "theta", t_float, d_theta), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "resolution_parameter", t_float, d_res), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "number_of_iterations", t_int, d_iters), "arg");
    Check(mgp_proc_add_result(sproc, kFieldNode, t_node), "res");
    Check(mgp_proc_add_result(sproc, kFieldCommunity, t_int), "res");
    Check(mgp_proc_add_result(sproc, kFieldCommunities, t_list_int), "res");
  } catch (const std::exception &) {
    if (d_wp) mgp_value_destroy(d_wp);
    if (d_gamma) mgp_value_destroy(d_gamma);
    if (d_theta) mgp_value_destroy(d_theta);
    if (d_res) mgp_value_destroy(d_res);
    if (d_iters) mgp_value_destroy(d_iters);
    return 1;
  }
  mgp_value_destroy(d_wp);
  mgp_value_destroy(d_gamma);
  mgp_value_destroy(d_theta);
  mgp_value_destroy(d_res);
  mgp_value_destroy(d_iters);
  return 0;
}

extern "C" int mgp_shutdown_module() { return 0; }
