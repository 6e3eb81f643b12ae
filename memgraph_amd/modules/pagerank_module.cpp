// pagerank.so — drop-in replacement for the reference MAGE pagerank module
// (src/mage/cpp/pagerank_module/pagerank_module.cpp), GPU-backed.
//
// Registered signature reproduced exactly (pagerank_module.cpp:122-136):
//   pagerank.get(max_iterations=100:int, damping_factor=0.85:float,
//                stop_epsilon=1e-5:float, num_of_threads=1:int)
//   -> (node: node, rank: float)
// num_of_threads is accepted for drop-in compatibility and ignored (the GPU
// sweep replaces the reference's edge-block thread pool,
// algorithm/pagerank.cpp:215-235).

#include "module_common.hpp"

namespace {

using namespace mgx_module;

constexpr const char *kProcedureGet = "get";
constexpr const char *kFieldNode = "node";
constexpr const char *kFieldRank = "rank";

void PagerankWrapper(mgp_list *args, mgp_graph *graph, mgp_result *result,
                     mgp_memory *memory) {
  try {
    const int64_t max_iterations = ArgInt(args, 0);
    const double damping_factor = ArgDouble(args, 1);
    const double stop_epsilon = ArgDouble(args, 2);
    (void)ArgInt(args, 3);  // num_of_threads: N/A on GPU

    // Scan (pagerank_module.cpp:18-54 semantics: scan-order dense ids).
    ScanResult scan = ScanGraph(graph, memory, Numbering::kVertexScanOrder);
    const int64_t V = (int64_t)scan.dense_to_mg.size();
    if (V == 0) return;

    mgx_context *ctx = Ctx();
    GraphGuard gg{ctx};
    CheckMgx(mgx_graph_from_coo(ctx, scan.src.data(), scan.dst.data(), nullptr, V,
                                (int64_t)scan.src.size(), MGX_BUILD_IN_CSR, &gg.g),
             "mgx_graph_from_coo");
    std::vector<double> rank(V);
    CheckMgx(mgx_pagerank(ctx, gg.g, max_iterations, damping_factor, stop_epsilon,
                          rank.data(), nullptr),
             "mgx_pagerank");

    for (int64_t v = 0; v < V; ++v) {
      EmitNodeRecord(graph, result, memory, scan.dense_to_mg[v], kFieldNode,
                     [&](mgp_result_record *rec) {
                       InsertDouble(rec, kFieldRank, rank[v], memory);
                     });
    }
  } catch (const std::exception &e) {
    // Never let an exception cross the ABI (pagerank_module.cpp:108-112).
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

}  // namespace

extern "C" int mgp_init_module(struct mgp_module *module, struct mgp_memory *memory) {
  mgp_value *d_max_iter = nullptr, *d_damping = nullptr, *d_eps = nullptr,
            *d_threads = nullptr;
  try {
    mgp_proc *proc = nullptr;
    Check(mgp_module_add_read_procedure(module, kProcedureGet, PagerankWrapper, &proc),
          "add_read_procedure");
    Check(mgp_value_make_int(100, memory, &d_max_iter), "make_int");
    Check(mgp_value_make_double(0.85, memory, &d_damping), "make_double");
    Check(mgp_value_make_double(1e-5, memory, &d_eps), "make_double");
    Check(mgp_value_make_int(1, memory, &d_threads), "make_int");

    mgp_type *t_int = nullptr, *t_float = nullptr, *t_node = nullptr;
    Check(mgp_type_int(&t_int), "type_int");
    Check(mgp_type_float(&t_float), "type_float");
    Check(mgp_type_node(&t_node), "type_node");

    Check(mgp_proc_add_opt_arg(proc, "max_iterations", t_int, d_max_iter), "opt_arg");
    Check(mgp_proc_add_opt_arg(proc, "damping_factor", t_float, d_damping), "opt_arg");
    Check(mgp_proc_add_opt_arg(proc, "stop_epsilon", t_float, d_eps), "opt_arg");
    Check(mgp_proc_add_opt_arg(proc, "num_of_threads", t_int, d_threads), "opt_arg");

    Check(mgp_proc_add_result(proc, kFieldNode, t_node), "add_result");
    Check(mgp_proc_add_result(proc, kFieldRank, t_float), "add_result");
  } catch (const std::exception &) {
    if (d_max_iter) mgp_value_destroy(d_max_iter);
    if (d_damping) mgp_value_destroy(d_damping);
    if (d_eps) mgp_value_destroy(d_eps);
    if (d_threads) mgp_value_destroy(d_threads);
    return 1;
  }
  mgp_value_destroy(d_max_iter);
  mgp_value_destroy(d_damping);
  mgp_value_destroy(d_eps);
  mgp_value_destroy(d_threads);
  return 0;
}

extern "C" int mgp_shutdown_module() { return 0; }
