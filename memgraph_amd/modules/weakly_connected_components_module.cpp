// weakly_connected_components.so — drop-in replacement for the reference
// connectivity module (src/mage/cpp/connectivity_module/
// connectivity_module.cpp), GPU-backed.
//
// Registered signature reproduced exactly (connectivity_module.cpp:91-99):
//   weakly_connected_components.get() -> (node: node, component_id: int)
// Component ids match the reference's BFS discovery order bit-exactly
// (min-member ascending == scan-order BFS roots).

#include "module_common.hpp"

namespace {

using namespace mgx_module;

constexpr const char *kProcedureGet = "get";
constexpr const char *kFieldNode = "node";
constexpr const char *kFieldComponentId = "component_id";

void Weak(mgp_list * /*args*/, mgp_graph *graph, mgp_result *result, mgp_memory *memory) {
  try {
    ScanResult scan = ScanGraph(graph, memory, Numbering::kVertexScanOrder);
    const int64_t V = (int64_t)scan.dense_to_mg.size();
    if (V == 0) return;

    mgx_context *ctx = Ctx();
    GraphGuard gg{ctx};
    CheckMgx(mgx_graph_from_coo(ctx, scan.src.data(), scan.dst.data(), nullptr, V,
                                (int64_t)scan.src.size(), MGX_BUILD_SYM_CSR, &gg.g),
             "mgx_graph_from_coo");
    std::vector<int64_t> component(V);
    int64_t n_components = 0;
    CheckMgx(mgx_wcc(ctx, gg.g, component.data(), &n_components), "mgx_wcc");

    for (int64_t v = 0; v < V; ++v) {
      EmitNodeRecord(graph, result, memory, scan.dense_to_mg[v], kFieldNode,
                     [&](mgp_result_record *rec) {
                       InsertInt(rec, kFieldComponentId, component[v], memory);
                     });
    }
  } catch (const std::exception &e) {
    (void)mgp_result_set_error_msg(result, e.what());
    return;
  }
}

}  // namespace

extern "C" int mgp_init_module(struct mgp_module *module, struct mgp_memory * /*memory*/) {
  try {
    mgp_proc *proc = nullptr;
    Check(mgp_module_add_read_procedure(module, kProcedureGet, Weak, &proc),
          "add_read_procedure");
    mgp_type *t_int = nullptr, *t_node = nullptr;
    Check(mgp_type_int(&t_int), "type_int");
    Check(mgp_type_node(&t_node), "type_node");
    Check(mgp_proc_add_result(proc, kFieldNode, t_node), "add_result");
    Check(mgp_proc_add_result(proc, kFieldComponentId, t_int), "add_result");
  } catch (const std::exception &) {
    return 1;
  }
  return 0;
}

extern "C" int mgp_shutdown_module() { return 0; }
