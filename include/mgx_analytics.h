// ============================================================================
// mgx_analytics — MI355X-native (gfx950/CDNA4) graph-analytics compute
// library: the GPU back end behind the drop-in MAGE module .so's
// (pagerank.so, katz_centrality.so, community_detection.so,
// weakly_connected_components.so).
//
// C ABI: plain pointers and sizes only. Each entry point cites the
// reference interface it replaces (paths under /root/reference).
// There is NO CPU fallback: every call fails with MGX_ERR_NO_DEVICE when no
// HIP device is present.
//
// Graphs are dense-id directed edge lists (ids in [0, n_vertices)); the
// module side performs the memgraph-id -> dense renumbering exactly as the
// reference modules do (pagerank_module.cpp:18-54, louvain.cpp:74-118).
// ============================================================================
#ifndef MGX_ANALYTICS_H
#define MGX_ANALYTICS_H

#include <stddef.h>
#include <stdint.h>

#ifdef __cplusplus
extern "C" {
#endif

typedef int32_t mgx_status;
#define MGX_OK 0
#define MGX_ERR_NO_DEVICE 1
#define MGX_ERR_HIP 2
#define MGX_ERR_INVALID_ARGUMENT 3
#define MGX_ERR_TOO_LARGE 4       /* V or E >= 2^31 */
#define MGX_ERR_OUT_OF_MEMORY 5
#define MGX_ERR_NCCL 6
#define MGX_ERR_NOT_SUPPORTED 7

const char *mgx_status_string(mgx_status s);
/* Human-readable detail of the most recent error on this thread. */
const char *mgx_last_error(void);

/* Number of visible HIP devices (0 => MGX_ERR_NO_DEVICE everywhere). */
int mgx_device_count(void);

typedef struct mgx_context mgx_context;  /* one device + stream + workspace */
typedef struct mgx_graph mgx_graph;      /* device-resident CSR container */

mgx_status mgx_init(int device, mgx_context **out);
mgx_status mgx_destroy(mgx_context *ctx);
/* Block until all queued work on the context stream is done. */
mgx_status mgx_sync(mgx_context *ctx);

/* ---- graph construction ------------------------------------------------ */

/* What to materialize on device. IN_CSR (row = destination, cols = sources)
 * feeds PageRank/Katz pull iterations; SYM_CSR (each edge twice, the layout
 * of louvain.cpp:158-233 in int32+fp32) feeds WCC/Louvain. */
#define MGX_BUILD_IN_CSR 1u
#define MGX_BUILD_SYM_CSR 2u
#define MGX_BUILD_WEIGHTED 4u /* with SYM_CSR: carry fp32 edge weights */
#define MGX_BUILD_OUT_CSR 8u /* row = source, cols = destinations (Brandes) */
#define MGX_BUILD_NO_PERM 16u /* identity vertex layout: skip the hot-first \
                               * renumbering (online/dynamic state keyed by \
                               * scan ids needs stable row ids across calls). \
                               * With IN_CSR|WEIGHTED also builds in_w. */

/* Upload a host COO and build the requested CSRs on device.
 * Replaces the layouts built by pagerank_alg::PageRankGraph
 * (algorithm/pagerank.cpp:166-182) and GetGrappoloSuitableGraph
 * (louvain.cpp:158-233). weights may be NULL (=> 1.0, the reference
 * default_weight). */
mgx_status mgx_graph_from_coo(mgx_context *ctx, const int64_t *src, const int64_t *dst,
                              const double *weights, int64_t n_vertices, int64_t n_edges,
                              uint32_t flags, mgx_graph **out);

/* Device-side deterministic generators (bit-identical to
 * include/mgx_graphgen.h / memgraph_amd/rmat.py / oracle). The COO never
 * leaves the device. weight_seed used only with MGX_BUILD_WEIGHTED. */
mgx_status mgx_graph_rmat(mgx_context *ctx, int scale, int64_t n_edges, uint64_t seed,
                          double a, double b, double c, uint32_t flags, uint64_t weight_seed,
                          mgx_graph **out);
mgx_status mgx_graph_uniform(mgx_context *ctx, int64_t n_vertices, int64_t n_edges,
                             uint64_t seed, uint32_t flags, uint64_t weight_seed,
                             mgx_graph **out);

/* Vertex-range sharded RMAT for multi-GPU PageRank: builds
 * the in-CSR of destination rows [row_begin, row_end) only, from the same
 * deterministic edge stream; out-degrees are global. */
mgx_status mgx_graph_rmat_sharded(mgx_context *ctx, int scale, int64_t n_edges, uint64_t seed,
                                  double a, double b, double c, int64_t row_begin,
                                  int64_t row_end, mgx_graph **out);

/* Test support: run the device generator and download the edge list, so
 * tests can check bit-identity with the numpy/C generators. */
mgx_status mgx_gen_edges_to_host(mgx_context *ctx, int rmat, int scale, int64_t n_vertices,
                                 int64_t n_edges, uint64_t seed, double a, double b,
                                 double c, int64_t *out_src, int64_t *out_dst);

mgx_status mgx_graph_destroy(mgx_context *ctx, mgx_graph *g);
int64_t mgx_graph_num_vertices(const mgx_graph *g);
int64_t mgx_graph_num_edges(const mgx_graph *g);
/* Edges held locally (== num_edges unless vertex-range sharded). */
int64_t mgx_graph_local_edges(const mgx_graph *g);
/* Device CSR build time (COO->CSR + degree bins), ms (the "CSR build ms"
 * component of BASELINE.json's metric). */
double mgx_graph_build_ms(const mgx_graph *g);

/* ---- PageRank (replaces pagerank_alg::ParallelIterativePageRank,
 *      algorithm/pagerank.cpp:194-242, and the pagerank.so module path) --- */

typedef struct {
  int64_t iterations;      /* iterations actually run */
  double iter_ms;          /* HIP-event time of all iteration sweeps */
  double sweep_ms;         /* dominant-kernel (fused SpMV sweep) total, ms */
  int64_t sweep_launches;  /* number of sweep launches (1/iteration) */
  double csr_build_ms;     /* copy of mgx_graph_build_ms */
  double download_ms;      /* normalize + f64 widen + D2H */
} mgx_pagerank_stats;

/* One-shot: init + iterate (reference stopping rule: stop when
 * Linf(new-old) <= stop_epsilon, checked each iteration, or at
 * max_iterations) + sum-normalize + download fp64 ranks.
 * out_rank may be NULL (bench discards values after warm checks).
 * stats may be NULL. */
mgx_status mgx_pagerank(mgx_context *ctx, mgx_graph *g, int64_t max_iterations,
                        double damping_factor, double stop_epsilon, double *out_rank,
                        mgx_pagerank_stats *stats);

/* Step-wise API for bench.py (time EXACTLY K iterations between syncs). */
typedef struct mgx_pagerank_run mgx_pagerank_run;
mgx_status mgx_pagerank_start(mgx_context *ctx, mgx_graph *g, double damping_factor,
                              mgx_pagerank_run **out);
/* Queue n iterations (async; no stopping rule, eps treated as 0). */
mgx_status mgx_pagerank_iterate(mgx_pagerank_run *run, int64_t n_iterations);
/* Iterate with the reference stopping rule (Linf(new-old) <= stop_epsilon,
 * pagerank.cpp:139-151), up to max_n iterations; *done = iterations run.
 * Synchronizes once per iteration to read the device Linf. On a
 * distributed run the Linf is max-allreduced first, so every rank runs the
 * same iteration count (no collective divergence). */
mgx_status mgx_pagerank_iterate_eps(mgx_pagerank_run *run, int64_t max_n,
                                    double stop_epsilon, int64_t *done);
/* Sweep timing accumulated so far (ms) and launch count. */
mgx_status mgx_pagerank_timing(mgx_pagerank_run *run, double *sweep_ms, int64_t *launches);
mgx_status mgx_pagerank_finish(mgx_pagerank_run *run, double *out_rank /* nullable */);

/* ---- WCC (replaces Weak, connectivity_module.cpp:41-87) ----------------
 * Component ids match the reference's BFS discovery order exactly
 * (min-member ascending). Needs SYM_CSR. */
mgx_status mgx_wcc(mgx_context *ctx, mgx_graph *g, int64_t *out_component,
                   int64_t *n_components);

/* ---- Katz (replaces katz_alg::SetKatz, katz.cpp:393-414) ---------------
 * Needs IN_CSR. Convergence rule replicates Converged (katz.cpp:165-215)
 * after the k-override, incl. divergent-series IEEE behavior. */
mgx_status mgx_katz(mgx_context *ctx, mgx_graph *g, double alpha, double epsilon,
                    double *out_centrality, int64_t *iterations);

/* ---- Louvain (replaces the grappolo basic path the module runs:
 *      runMultiPhaseBasic.cpp:53-146 + parallelLouvainMethod.cpp:65-290)
 * Needs SYM_CSR (weighted or unweighted). Vertices with no edges get
 * community -1 exactly when the reference would. */
mgx_status mgx_louvain(mgx_context *ctx, mgx_graph *g, double threshold,
                       int64_t *out_community, int64_t *n_communities);

/* ---- Betweenness centrality (replaces
 *      betweenness_centrality_alg::BetweennessCentrality,
 *      betweenness_centrality_module/algorithm/betweenness_centrality.cpp
 *      :71-147 — exact Brandes, O(V*E)) -----------------------------------
 * directed needs MGX_BUILD_OUT_CSR; undirected needs MGX_BUILD_SYM_CSR. */
mgx_status mgx_betweenness(mgx_context *ctx, mgx_graph *g, int directed, int normalize,
                           double *out_bc);

/* ---- Online (dynamic) PageRank — replaces pagerank_online_alg
 *      (query_modules/pagerank_module/algorithm_online/pagerank.cpp:253-318:
 *      R random walks per node, stop probability epsilon; rank[v] =
 *      visit_count[v] / sum; updates truncate walks at the changed node and
 *      regrow the suffix with epsilon/2) ---------------------------------
 * State is per-process (the reference keeps a global context, pagerank.cpp
 * :49). Walks live on device: an entry pool of (walk, pos, node-slot)
 * triples plus per-walk start/liveness; counters are recomputed from live
 * entries (the reference's walks vector is equally the ground truth, its
 * counter map a cache). Nodes are keyed by memgraph id via a host slot map
 * so state survives graph changes between calls.
 * RNG is counter-based (splitmix64 streams keyed by walk/generation/step)
 * seeded from `seed`; the reference seeds from std::random_device, so
 * parity is the statistical bar in DESIGN.md, not bit-equality.
 * Graph must be built with MGX_BUILD_OUT_CSR; dense_to_mg maps the scan's
 * dense ids to memgraph ids. out_rank is by dense id. */
mgx_status mgx_pronline_set(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                            int64_t R, double eps, uint64_t seed, double *out_rank);
/* consistent=0 <=> some scanned node has no walk state (caller must raise
 * the reference's inconsistency error, pagerank.cpp:284-288). */
mgx_status mgx_pronline_get(mgx_context *ctx, const int64_t *dense_to_mg, int64_t V,
                            double *out_rank, int *consistent);
mgx_status mgx_pronline_update(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                               const int64_t *created_v, int64_t n_cv,
                               const int64_t *created_e /* (from,to) mg-id pairs */,
                               int64_t n_ce, const int64_t *deleted_v, int64_t n_dv,
                               const int64_t *deleted_e, int64_t n_de, double *out_rank);
mgx_status mgx_pronline_reset(mgx_context *ctx);
int mgx_pronline_initialized(void);
/* Test support: walk-state invariants (counts by live entries). */
mgx_status mgx_pronline_stats(mgx_context *ctx, int64_t *n_walks, int64_t *n_live_walks,
                              int64_t *n_live_entries);

/* ---- Online (dynamic) Katz centrality — replaces the reference's online
 *      katz_alg (query_modules/katz_centrality_module/algorithm/katz.cpp:
 *      SetKatz :356-378, UpdateKatz :380-468, KatzCentralityLoop :211-240).
 * Deterministic; parity vs the sequential oracle at 1e-9 (f64 sum order).
 * State is per-process like the reference's context (katz.cpp:105).
 * Graph needs MGX_BUILD_IN_CSR|MGX_BUILD_OUT_CSR. Created edges arrive as
 * (from,to) mg-id pairs; skipping by pair multiplicity is value-equivalent
 * to the reference's skip-by-edge-id (see csrc/katz_online.hip header). */
mgx_status mgx_konline_set(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                           double alpha, double epsilon, double *out /* [V] */);
mgx_status mgx_konline_get(mgx_context *ctx, const int64_t *dense_to_mg, int64_t V,
                           double *out, int *consistent);
mgx_status mgx_konline_update(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                              const int64_t *created_v, int64_t n_cv,
                              const int64_t *created_e, int64_t n_ce,
                              const int64_t *deleted_v, int64_t n_dv,
                              const int64_t *deleted_e, int64_t n_de, double *out);
mgx_status mgx_konline_reset(mgx_context *ctx);
int mgx_konline_initialized(void);
int64_t mgx_konline_iterations(void);

/* ---- LabelRankT online community detection — replaces LabelRankT
 *      (query_modules/community_detection_module/algorithm_online/
 *      community_detection.cpp; SetLabels :311-328, UpdateLabels :330-351,
 *      GetLabels :305-309). Deterministic; parity = identical labels vs the
 *      sequential oracle (pinned against the compiled reference core).
 * Graph: undirected -> MGX_BUILD_SYM_CSR (|WEIGHTED); directed ->
 * MGX_BUILD_IN_CSR|MGX_BUILD_NO_PERM (|WEIGHTED). out_label is by dense id,
 * renumbered 1..k as AllLabels does (-1 = unlabeled). */
mgx_status mgx_lrt_set(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                       int directed, int weighted, double similarity_threshold,
                       double exponent, double min_value, double w_selfloop,
                       int64_t max_iterations, int64_t max_updates, int64_t *out_label);
/* ran_set=1 <=> state was uncalculated and a full (non-persisted) compute
 * ran, as GetLabels does. */
mgx_status mgx_lrt_get(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                       int64_t *out_label, int *ran_set);
mgx_status mgx_lrt_update(mgx_context *ctx, mgx_graph *g, const int64_t *dense_to_mg,
                          const int64_t *mod_v, int64_t n_mv, const int64_t *mod_e,
                          int64_t n_me, const int64_t *del_v, int64_t n_dv,
                          const int64_t *del_e, int64_t n_de, int64_t *out_label);
mgx_status mgx_lrt_reset(mgx_context *ctx);
int mgx_lrt_initialized(void);

/* ---- Leiden community detection — replaces leiden_alg::GetCommunities
 *      (src/mage/cpp/leiden_community_detection_module/algorithm/
 *      leiden.cpp:569-591; CPM local moves, probabilistic refinement,
 *      first-edge aggregation). RANDOMIZED in the reference
 *      (random_device-seeded shuffle/draws), so parity is the DESIGN.md
 *      statistical bar; `seed` pins our counter-based streams.
 * Graph needs MGX_BUILD_SYM_CSR (|WEIGHTED). out_hier[v*cap+k] = the
 * community id of v at dendrogram level k (-1 padded); out_levels[v] =
 * hierarchy depth (0 when the graph has no edges — the module then raises
 * the reference's "No communities detected." error path). */
mgx_status mgx_leiden(mgx_context *ctx, mgx_graph *g, double gamma, double theta,
                      double resolution, int64_t max_iterations, uint64_t seed,
                      int64_t cap, int64_t *out_hier, int64_t *out_levels);

/* ---- multi-GPU (RCCL over xGMI) ------------------------- */

#define MGX_UNIQUE_ID_BYTES 128 /* == sizeof(ncclUniqueId) */
mgx_status mgx_comm_unique_id(void *out_bytes /* [MGX_UNIQUE_ID_BYTES] */);
mgx_status mgx_comm_init(mgx_context *ctx, int rank, int world_size, const void *id_bytes);
mgx_status mgx_comm_destroy(mgx_context *ctx);

/* Sharded PageRank run: graph must be a sharded in-CSR (rows
 * [row_begin,row_end)); each iteration all-gathers the owned contrib/rank
 * slices (ncclAllGather over xGMI). Same timing semantics as the
 * single-GPU run API. */
mgx_status mgx_pagerank_start_dist(mgx_context *ctx, mgx_graph *g, double damping_factor,
                                   int64_t row_begin, int64_t row_end,
                                   mgx_pagerank_run **out);

#ifdef __cplusplus
}
#endif

#endif /* MGX_ANALYTICS_H */
