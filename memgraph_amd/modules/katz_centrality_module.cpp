// katz_centrality.so — drop-in replacement for the reference MAGE katz
// module (src/mage/cpp/katz_centrality_module/katz_centrality_module.cpp),
// GPU-backed.
//
// Registered signature reproduced exactly (katz_centrality_module.cpp:57-74):
//   katz_centrality.get(alpha=0.2:float, epsilon=1e-2:float)
//   -> (node: node, rank: float)
// (The reference's double-destroy of default_alpha and leak of
// default_epsilon at :72-73 are NOT replicated — SURVEY.md appendix 7.)

#include "module_common.hpp"

namespace {

using namespace mgx_module;

constexpr const char *kProcedureGet = "get";
constexpr const char *kFieldNode = "node";
constexpr const char *kFieldRank = "rank";

void GetKatzCentrality(mgp_list *args, mgp_graph *graph, mgp_result *result,
                       mgp_memory *memory) {
  try {
    const double alpha = ArgDouble(args, 0);
    const double epsilon = ArgDouble(args, 1);

    ScanResult scan = ScanGraph(graph, memory, Numbering::kVertexScanOrder);
    const int64_t V = (int64_t)scan.dense_to_mg.size();
    if (V == 0) return;

    mgx_context *ctx = Ctx();
    GraphGuard gg{ctx};
    CheckMgx(mgx_graph_from_coo(ctx, scan.src.data(), scan.dst.data(), nullptr, V,
                                (int64_t)scan.src.size(), MGX_BUILD_IN_CSR, &gg.g),
             "mgx_graph_from_coo");
    std::vector<double> centrality(V);
    int64_t iters = 0;
    CheckMgx(mgx_katz(ctx, gg.g, alpha, epsilon, centrality.data(), &iters), "mgx_katz");

    for (int64_t v = 0; v < V; ++v) {
      EmitNodeRecord(graph, result, memory, scan.dense_to_mg[v], kFieldNode,
                     [&](mgp_result_record *rec) {
                       InsertDouble(rec, kFieldRank, centrality[v], memory);
                     });
    }
  } catch (const std::exception &e) {
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

}  // namespace

extern "C" int mgp_init_module(struct mgp_module *module, struct mgp_memory *memory) {
  mgp_value *d_alpha = nullptr, *d_eps = nullptr;
  try {
    mgp_proc *proc = nullptr;
    Check(mgp_module_add_read_procedure(module, kProcedureGet, GetKatzCentrality, &proc),
          "add_read_procedure");
    Check(mgp_value_make_double(0.2, memory, &d_alpha), "make_double");
    Check(mgp_value_make_double(1e-2, memory, &d_eps), "make_double");

    mgp_type *t_float = nullptr, *t_node = nullptr;
    Check(mgp_type_float(&t_float), "type_float");
    Check(mgp_type_node(&t_node), "type_node");

    Check(mgp_proc_add_opt_arg(proc, "alpha", t_float, d_alpha), "opt_arg");
    Check(mgp_proc_add_opt_arg(proc, "epsilon", t_float, d_eps), "opt_arg");

    Check(mgp_proc_add_result(proc, kFieldNode, t_node), "add_result");
    Check(mgp_proc_add_result(proc, kFieldRank, t_float), "add_result");
  } catch (const std::exception &) {
    if (d_alpha) mgp_value_destroy(d_alpha);
    if (d_eps) mgp_value_destroy(d_eps);
    return 1;
  }
  mgp_value_destroy(d_alpha);
  mgp_value_destroy(d_eps);
  return 0;
}

extern "C" int mgp_shutdown_module() { return 0; }
