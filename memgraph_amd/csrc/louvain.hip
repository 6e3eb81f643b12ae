// Louvain community detection on gfx950 — replaces the grappolo basic path
// the reference community_detection module runs (runMultiPhaseBasic.cpp
// :53-146 + parallelLouvainMethod.cpp:65-290). Implementation lands in this
// round after the SpMV-family kernels are validated; until then the entry
// point reports NOT_SUPPORTED (never a CPU fallback).

#include "mgx_internal.h"

mgx_status mgx_louvain_impl(mgx_context *ctx, mgx_graph *g, double threshold,
                            int64_t *out_community, int64_t *n_communities) {
  (void)ctx;
  (void)g;
  (void)threshold;
  (void)out_community;
  (void)n_communities;
  mgx_set_error("mgx_louvain: GPU implementation pending (round 1 WIP)");
  return MGX_ERR_NOT_SUPPORTED;
}
