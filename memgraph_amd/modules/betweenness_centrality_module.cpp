// betweenness_centrality.so — drop-in replacement for the reference MAGE
// betweenness module (src/mage/cpp/betweenness_centrality_module/
// betweenness_centrality_module.cpp), GPU-backed exact Brandes.
//
// Registered signature reproduced exactly (betweenness_centrality_module.cpp
// :77-90): betweenness_centrality.get(directed=true:bool,
// normalized=true:bool, threads=<hardware_concurrency>:int)
// -> (node: node, betweenness_centrality: float).
// threads is accepted for drop-in compatibility and ignored (the batched
// GPU Brandes replaces the reference's std::async source partitioning).

#include <thread>

#include "module_common.hpp"

namespace {

using namespace mgx_module;

constexpr const char *kProcedureGet = "get";
constexpr const char *kFieldNode = "node";
constexpr const char *kFieldBCScore = "betweenness_centrality";

int64_t ArgBool(mgp_list *args, size_t i) {
  mgp_value *v = nullptr;
  Check(mgp_list_at(args, i, &v), "list_at");
  int out = 0;
  Check(mgp_value_get_bool(v, &out), "value_get_bool");
  return out;
}

void GetBetweennessCentrality(mgp_list *args, mgp_graph *graph, mgp_result *result,
                              mgp_memory *memory) {
  try {
    const bool directed = ArgBool(args, 0) != 0;
    const bool normalize = ArgBool(args, 1) != 0;
    (void)ArgInt(args, 2);  // threads: N/A on GPU

    ScanResult scan = ScanGraph(graph, memory, Numbering::kVertexScanOrder);
    const int64_t V = (int64_t)scan.dense_to_mg.size();
    if (V == 0) return;

    mgx_context *ctx = Ctx();
    GraphGuard gg{ctx};
    CheckMgx(mgx_graph_from_coo(ctx, scan.src.data(), scan.dst.data(), nullptr, V,
                                (int64_t)scan.src.size(),
                                directed ? MGX_BUILD_OUT_CSR : MGX_BUILD_SYM_CSR, &gg.g),
             "mgx_graph_from_coo");
    std::vector<double> bc(V);
    CheckMgx(mgx_betweenness(ctx, gg.g, directed ? 1 : 0, normalize ? 1 : 0, bc.data()),
             "mgx_betweenness");

    for (int64_t v = 0; v < V; ++v) {
      EmitNodeRecord(graph, result, memory, scan.dense_to_mg[v], kFieldNode,
                     [&](mgp_result_record *rec) {
                       InsertDouble(rec, kFieldBCScore, bc[v], memory);
                     });
    }
  } catch (const std::exception &e) {
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

}  // namespace

extern "C" int mgp_init_module(struct mgp_module *module, struct mgp_memory *memory) {
  mgp_value *d_directed = nullptr, *d_normalized = nullptr, *d_threads = nullptr;
  try {
    mgp_proc *proc = nullptr;
    Check(mgp_module_add_read_procedure(module, kProcedureGet, GetBetweennessCentrality,
                                        &proc),
          "add_read_procedure");
    Check(mgp_value_make_bool(1, memory, &d_directed), "make_bool");
    Check(mgp_value_make_bool(1, memory, &d_normalized), "make_bool");
    Check(mgp_value_make_int((int64_t)std::thread::hardware_concurrency(), memory,
                             &d_threads),
          "make_int");

    mgp_type *t_bool = nullptr, *t_int = nullptr, *t_float = nullptr, *t_node = nullptr;
    Check(mgp_type_bool(&t_bool), "type_bool");
    Check(mgp_type_int(&t_int), "type_int");
    Check(mgp_type_float(&t_float), "type_float");
    Check(mgp_type_node(&t_node), "type_node");

    Check(mgp_proc_add_opt_arg(proc, "directed", t_bool, d_directed), "opt_arg");
    Check(mgp_proc_add_opt_arg(proc, "normalized", t_bool, d_normalized), "opt_arg");
    Check(mgp_proc_add_opt_arg(proc, "threads", t_int, d_threads), "opt_arg");

    Check(mgp_proc_add_result(proc, kFieldNode, t_node), "add_result");
    Check(mgp_proc_add_result(proc, kFieldBCScore, t_float), "add_result");
  } catch (const std::exception &) {
    if (d_directed) mgp_value_destroy(d_directed);
    if (d_normalized) mgp_value_destroy(d_normalized);
    if (d_threads) mgp_value_destroy(d_threads);
    return 1;
  }
  mgp_value_destroy(d_directed);
  mgp_value_destroy(d_normalized);
  mgp_value_destroy(d_threads);
  return 0;
}

extern "C" int mgp_shutdown_module() { return 0; }
