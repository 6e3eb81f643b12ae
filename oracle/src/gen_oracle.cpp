// ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
// Deterministic generators; bit-identical to include/mgx_graphgen.h users.

#include "../../include/mgx_graphgen.h"
#include "../oracle.h"

extern "C" void oracle_gen_rmat(int64_t scale, int64_t n_edges, uint64_t seed, double a,
                                double b, double c, int64_t *out_src, int64_t *out_dst) {
  const uint64_t ms = mgx_seed_mix(seed);
  const mgx_rmat_thresholds t = mgx_rmat_make_thresholds(a, b, c);
  for (int64_t i = 0; i < n_edges; ++i) {
    uint64_t s, d;
    mgx_rmat_edge(ms, (uint64_t)i, (int)scale, t, &s, &d);
    out_src[i] = (int64_t)s;
    out_dst[i] = (int64_t)d;
  }
}

extern "C" void oracle_gen_uniform(int64_t n_vertices, int64_t n_edges, uint64_t seed,
                                   int64_t *out_src, int64_t *out_dst) {
  const uint64_t ms = mgx_seed_mix(seed);
  for (int64_t i = 0; i < n_edges; ++i) {
    uint64_t s, d;
    mgx_uniform_edge(ms, (uint64_t)i, (uint64_t)n_vertices, &s, &d);
    out_src[i] = (int64_t)s;
    out_dst[i] = (int64_t)d;
  }
}

extern "C" void oracle_gen_weights(int64_t n_edges, uint64_t seed, double *out_w) {
  const uint64_t ms = mgx_seed_mix(seed);
  for (int64_t i = 0; i < n_edges; ++i) out_w[i] = mgx_edge_weight(ms, (uint64_t)i);
}
