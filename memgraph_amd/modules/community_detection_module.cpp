// community_detection.so — drop-in replacement for the reference MAGE
// community detection module (src/mage/cpp/community_detection_module/
// community_detection_module.cpp), GPU-backed Louvain.
//
// Registered signature reproduced exactly (community_detection_module.cpp:111-136):
//   community_detection.get(weight_property="weight":string,
//                           coloring=false:bool, min_graph_shrink=100000:int,
//                           community_alg_threshold=1e-6:float,
//                           coloring_alg_threshold=0.01:float,
//                           num_of_threads=<cores/2>:int)
//   -> (node: node, community_id: int)
// plus community_detection.get_subgraph(subgraph_nodes: list<node>,
// subgraph_relationships: list<relationship>, <same optional args>)
// (community_detection_module.cpp:136-152 / OnSubgraph).
// min_graph_shrink / coloring_alg_threshold / num_of_threads are accepted
// for drop-in compatibility; minGraphSize is unused by the reference's basic
// path too (runMultiPhaseBasic ignores it), and thread count is N/A on GPU.
// coloring=true is REJECTED with an explicit error (the reference's coloring
// path is a different, coloring-scheduled algorithm — silent substitution of
// the basic one would be a behavioral divergence).

#include <thread>

#include "module_common.hpp"

namespace {

using namespace mgx_module;

constexpr const char *kProcedureGet = "get";
constexpr const char *kFieldNode = "node";
constexpr const char *kFieldCommunity = "community_id";
constexpr const char *kDefaultWeightProperty = "weight";
constexpr double kDefaultWeight = 1.0;

void RunLouvain(mgp_graph *graph, mgp_result *result, mgp_memory *memory,
                const ScanResult &scan, double threshold) {
  const int64_t V = (int64_t)scan.dense_to_mg.size();
  // The reference emits nothing when the scanned graph has no edges
  // (community_detection_module.cpp:72-74).
  if (V == 0 || scan.src.empty()) return;

  mgx_context *ctx = Ctx();
  GraphGuard gg{ctx};
  CheckMgx(mgx_graph_from_coo(ctx, scan.src.data(), scan.dst.data(),
                              scan.weights.data(), V, (int64_t)scan.src.size(),
                              MGX_BUILD_SYM_CSR | MGX_BUILD_WEIGHTED, &gg.g),
           "mgx_graph_from_coo");
  std::vector<int64_t> community(V);
  int64_t n_communities = 0;
  CheckMgx(mgx_louvain(ctx, gg.g, threshold, community.data(), &n_communities),
           "mgx_louvain");

  for (int64_t v = 0; v < V; ++v) {
    EmitNodeRecord(graph, result, memory, scan.dense_to_mg[v], kFieldNode,
                   [&](mgp_result_record *rec) {
                     InsertInt(rec, kFieldCommunity, community[v], memory);
                   });
  }
}

void OnGraph(mgp_list *args, mgp_graph *graph, mgp_result *result, mgp_memory *memory) {
  try {
    const char *weight_property = ArgString(args, 0);
    // args 2,4,5 accepted, see header comment.
    const bool coloring = ArgBool(args, 1);
    const double threshold = ArgDouble(args, 3);
    if (coloring) {
      // Honest rejection instead of silently running the basic algorithm:
      // the reference's coloring=true selects grappolo's runMultiPhaseColoring
      // (louvain.cpp:42-48), a different (coloring-scheduled Gauss-Seidel)
      // variant whose partitions differ from the basic path's.
      throw std::runtime_error(
          "coloring=true is not supported by the GPU backend; "
          "call community_detection.get(coloring=false)");
    }

    // Louvain numbering: first-seen dense ids (louvain.cpp:86-117) — the
    // observable community numbering depends on it.
    ScanResult scan = ScanGraph(graph, memory, Numbering::kFirstSeen,
                                /*read_weights=*/true, weight_property, kDefaultWeight);
    RunLouvain(graph, result, memory, scan, threshold);
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
    // Argument layout of the reference's get_subgraph
    // (LouvainCommunityDetection with subgraph=true: the two lists first,
    // then the same optionals — community_detection_module.cpp:51-64).
    mgp_list *nodes = ArgList(args, 0);
    mgp_list *relationships = ArgList(args, 1);
    const char *weight_property = ArgString(args, 2);
    const bool coloring = ArgBool(args, 3);
    const double threshold = ArgDouble(args, 5);
    if (coloring) {
      throw std::runtime_error(
          "coloring=true is not supported by the GPU backend; "
          "call community_detection.get_subgraph(..., coloring=false)");
    }
    ScanResult scan = ScanSubgraph(graph, memory, nodes, relationships,
                                   /*read_weights=*/true, weight_property,
                                   kDefaultWeight);
    RunLouvain(graph, result, memory, scan, threshold);
  } catch (const std::exception &e) {
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

}  // namespace

extern "C" int mgp_init_module(struct mgp_module *module, struct mgp_memory *memory) {
  mgp_value *d_weight_prop = nullptr, *d_coloring = nullptr, *d_shrink = nullptr,
            *d_threshold = nullptr, *d_col_threshold = nullptr, *d_threads = nullptr;
  try {
    mgp_proc *proc = nullptr;
    Check(mgp_module_add_read_procedure(module, kProcedureGet, OnGraph, &proc),
          "add_read_procedure");
    const int64_t default_threads =
        (int64_t)(std::thread::hardware_concurrency() / 2);
    Check(mgp_value_make_string(kDefaultWeightProperty, memory, &d_weight_prop), "mk");
    Check(mgp_value_make_bool(0, memory, &d_coloring), "mk");
    Check(mgp_value_make_int(100000, memory, &d_shrink), "mk");
    Check(mgp_value_make_double(1e-6, memory, &d_threshold), "mk");
    Check(mgp_value_make_double(0.01, memory, &d_col_threshold), "mk");
    Check(mgp_value_make_int(default_threads, memory, &d_threads), "mk");

    mgp_type *t_int = nullptr, *t_float = nullptr, *t_bool = nullptr, *t_string = nullptr,
             *t_node = nullptr;
    Check(mgp_type_int(&t_int), "type_int");
    Check(mgp_type_float(&t_float), "type_float");
    Check(mgp_type_bool(&t_bool), "type_bool");
    Check(mgp_type_string(&t_string), "type_string");
    Check(mgp_type_node(&t_node), "type_node");

    Check(mgp_proc_add_opt_arg(proc, "weight_property", t_string, d_weight_prop), "arg");
    Check(mgp_proc_add_opt_arg(proc, "coloring", t_bool, d_coloring), "arg");
    Check(mgp_proc_add_opt_arg(proc, "min_graph_shrink", t_int, d_shrink), "arg");
    Check(mgp_proc_add_opt_arg(proc, "community_alg_threshold", t_float, d_threshold),
          "arg");
    Check(mgp_proc_add_opt_arg(proc, "coloring_alg_threshold", t_float, d_col_threshold),
          "arg");
    Check(mgp_proc_add_opt_arg(proc, "num_of_threads", t_int, d_threads), "arg");

    Check(mgp_proc_add_result(proc, kFieldNode, t_node), "add_result");
    Check(mgp_proc_add_result(proc, kFieldCommunity, t_int), "add_result");

    // get_subgraph (community_detection_module.cpp:136-152)
    mgp_proc *sproc = nullptr;
    Check(mgp_module_add_read_procedure(module, "get_subgraph", OnSubgraph, &sproc),
          "add_read_procedure(get_subgraph)");
    mgp_type *t_rel = nullptr, *t_list_node = nullptr, *t_list_rel = nullptr;
    Check(mgp_type_relationship(&t_rel), "type_relationship");
    Check(mgp_type_list(t_node, &t_list_node), "type_list(node)");
    Check(mgp_type_list(t_rel, &t_list_rel), "type_list(rel)");
    Check(mgp_proc_add_arg(sproc, "subgraph_nodes", t_list_node), "arg");
    Check(mgp_proc_add_arg(sproc, "subgraph_relationships", t_list_rel), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "weight_property", t_string, d_weight_prop), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "coloring", t_bool, d_coloring), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "min_graph_shrink", t_int, d_shrink), "arg");
    Check(mgp_proc_add_opt_arg(sproc, "community_alg_threshold", t_float, d_threshold),
          "arg");
    Check(mgp_proc_add_opt_arg(sproc, "coloring_alg_threshold", t_float, d_col_threshold),
          "arg");
    Check(mgp_proc_add_opt_arg(sproc, "num_of_threads", t_int, d_threads), "arg");
    Check(mgp_proc_add_result(sproc, kFieldNode, t_node), "add_result");
    Check(mgp_proc_add_result(sproc, kFieldCommunity, t_int), "add_result");
  } catch (const std::exception &) {
    if (d_weight_prop) mgp_value_destroy(d_weight_prop);
    if (d_coloring) mgp_value_destroy(d_coloring);
    if (d_shrink) mgp_value_destroy(d_shrink);
    if (d_threshold) mgp_value_destroy(d_threshold);
    if (d_col_threshold) mgp_value_destroy(d_col_threshold);
    if (d_threads) mgp_value_destroy(d_threads);
    return 1;
  }
  mgp_value_destroy(d_weight_prop);
  mgp_value_destroy(d_coloring);
  mgp_value_destroy(d_shrink);
  mgp_value_destroy(d_threshold);
  mgp_value_destroy(d_col_threshold);
  mgp_value_destroy(d_threads);
  return 0;
}

extern "C" int mgp_shutdown_module() { return 0; }
