// PageRank on gfx950 — replaces pagerank_alg::ParallelIterativePageRank
// (reference algorithm/pagerank.cpp:194-242) with a fused pull-SpMV sweep:
//
//   new_rank[v] = (1-d)/N + d * sum_{u->v} contrib[u],  contrib[u] = rank[u]/outdeg(u)
//
// One kernel launch per iteration covers all degree bins (sub-wave work
// assignment per mgx_bins): f32 ranks/contrib in HBM, f64 accumulation in
// registers, and the same kernel writes new_rank AND new_contrib and folds
// the per-block Linf(delta) into a device scalar — so per-iteration traffic
// is the algorithmic 8 B/edge + ~20 B/vertex (DESIGN.md roofline).
//
// Parity bar (tests/test_gpu_pagerank.py): |r_gpu - r_cpu|inf <= 1e-6 vs the
// fp64 oracle after equal iterations, post sum-normalize.

#include <chrono>

#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;

struct PrArgs {
  const uint32_t *row_ptr;
  const int32_t *col;
  const int32_t *bin_rows;
  int64_t n[4];     // rows per bin
  int64_t off[4];   // offsets into bin_rows
  int64_t goff[4];  // grid offsets per section
  int64_t grid[4];
  const float *contrib_old;
  const float *rank_old;
  float *rank_new;
  float *contrib_new;
  const float *inv_outdeg;  // global vertex id
  int64_t row_base;         // sharded: global row = row_base + local row
  float base_term;          // (1-d)/N
  float damping;
  uint32_t *delta_max;      // f32-as-ordered-uint
  // Source-striped sweep (large V): per-stripe row sub-ranges
  // and fp64 row partials. mode: 0 = direct (single launch),
  // 1 = first stripe (partial = acc), 2 = middle (partial += acc),
  // 3 = last stripe (finish from partial + acc).
  const uint32_t *sp_lo;
  const uint32_t *sp_hi;
  double *partial;
  int mode;
  int deep;  // experiment: 4x-unrolled wide path (16 gathers/lane in flight)
};

template <int LANES>
__device__ inline float pr_rows(const PrArgs &A, int sec, int64_t block_in_sec) {
  constexpr int RPB = kBlock / LANES;
  const int64_t nrows = A.n[sec];
  const int32_t *rows_list = A.bin_rows + A.off[sec];
  const int sub = threadIdx.x % LANES;
  float maxd = 0.0f;
  __shared__ double red[4];
  for (int64_t base = block_in_sec * RPB; base < nrows; base += A.grid[sec] * RPB) {
    const int64_t ri = base + threadIdx.x / LANES;
    double acc = 0.0;
    int32_t row = -1;
    if (ri < nrows) {
      row = rows_list[ri];
      const uint32_t s = A.sp_lo[row], e = A.sp_hi[row];
      if constexpr (LANES >= 64) {
        // Wide rows: scalar head to 16-B alignment, then int4 nontemporal
        // column loads (the col stream is read exactly once — keep it out
        // of L1 so the contrib gathers stay cached; G13/nt-weights) with 4
        // independent gathers per lane per iteration for latency hiding.
        uint32_t s_al = (s + 3u) & ~3u;
        if (s_al > e) s_al = e;
        for (uint32_t j = s + sub; j < s_al; j += LANES)
          acc += (double)A.contrib_old[A.col[j]];
        const uint32_t nvec = (e - s_al) / 4;
        typedef int v4i __attribute__((ext_vector_type(4)));
        const v4i *col4 = reinterpret_cast<const v4i *>(A.col + s_al);
        uint32_t c = sub;
        if (A.deep) {
          // 4x unrolled: 16 independent gathers in flight per lane
          // (MGX_PR_DEEP_UNROLL experiment).
          for (; c + 3 * LANES < nvec; c += 4 * LANES) {
            const v4i c0 = __builtin_nontemporal_load(col4 + c);
            const v4i c1 = __builtin_nontemporal_load(col4 + c + LANES);
            const v4i c2 = __builtin_nontemporal_load(col4 + c + 2 * LANES);
            const v4i c3 = __builtin_nontemporal_load(col4 + c + 3 * LANES);
            float g[16];
            g[0] = A.contrib_old[c0.x]; g[1] = A.contrib_old[c0.y];
            g[2] = A.contrib_old[c0.z]; g[3] = A.contrib_old[c0.w];
            g[4] = A.contrib_old[c1.x]; g[5] = A.contrib_old[c1.y];
            g[6] = A.contrib_old[c1.z]; g[7] = A.contrib_old[c1.w];
            g[8] = A.contrib_old[c2.x]; g[9] = A.contrib_old[c2.y];
            g[10] = A.contrib_old[c2.z]; g[11] = A.contrib_old[c2.w];
            g[12] = A.contrib_old[c3.x]; g[13] = A.contrib_old[c3.y];
            g[14] = A.contrib_old[c3.z]; g[15] = A.contrib_old[c3.w];
            for (int k = 0; k < 16; ++k) acc += (double)g[k];
          }
        }
        // 2x unrolled: 8 independent gathers in flight per lane.
        for (; c + LANES < nvec; c += 2 * LANES) {
          const v4i c0 = __builtin_nontemporal_load(col4 + c);
          const v4i c1 = __builtin_nontemporal_load(col4 + c + LANES);
          const float g0 = A.contrib_old[c0.x];
          const float g1 = A.contrib_old[c0.y];
          const float g2 = A.contrib_old[c0.z];
          const float g3 = A.contrib_old[c0.w];
          const float g4 = A.contrib_old[c1.x];
          const float g5 = A.contrib_old[c1.y];
          const float g6 = A.contrib_old[c1.z];
          const float g7 = A.contrib_old[c1.w];
          acc += (double)g0;
          acc += (double)g1;
          acc += (double)g2;
          acc += (double)g3;
          acc += (double)g4;
          acc += (double)g5;
          acc += (double)g6;
          acc += (double)g7;
        }
        for (; c < nvec; c += LANES) {
          const v4i cc = __builtin_nontemporal_load(col4 + c);
          acc += (double)A.contrib_old[cc.x];
          acc += (double)A.contrib_old[cc.y];
          acc += (double)A.contrib_old[cc.z];
          acc += (double)A.contrib_old[cc.w];
        }
        for (uint32_t j = s_al + nvec * 4 + sub; j < e; j += LANES)
          acc += (double)A.contrib_old[A.col[j]];
      } else {
        // Short rows: a LANES-wide group reads consecutive cols (16/64-B
        // granules); nt keeps the one-pass col stream out of L1.
        for (uint32_t j = s + sub; j < e; j += LANES)
          acc += (double)A.contrib_old[__builtin_nontemporal_load(A.col + j)];
      }
    }
    if constexpr (LANES <= 64) {
      for (int o = LANES / 2; o; o >>= 1) acc += __shfl_down(acc, o, LANES);
    } else {
      for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
      if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
      __syncthreads();
      if (threadIdx.x == 0) acc = red[0] + red[1] + red[2] + red[3];
    }
    if (sub == 0 && row >= 0 && (LANES <= 64 || threadIdx.x == 0)) {
      if (A.mode == 1) {
        A.partial[row] = acc;
      } else if (A.mode == 2) {
        A.partial[row] += acc;
      } else {
        if (A.mode == 3) acc += A.partial[row];
        const float newr = (float)((double)A.base_term + (double)A.damping * acc);
        const int64_t gv = A.row_base + row;
        A.rank_new[gv] = newr;
        A.contrib_new[gv] = newr * A.inv_outdeg[gv];
        if (A.delta_max) {  // old-rank read only paid when Linf is tracked
          const float d = fabsf(newr - A.rank_old[gv]);
          if (d > maxd) maxd = d;
        }
      }
    }
    if constexpr (LANES > 64) __syncthreads();
  }
  return maxd;
}

__global__ void __launch_bounds__(kBlock) k_pr_sweep(PrArgs A) {
  const int64_t b = blockIdx.x;
  int sec = 3;
  if (b < A.goff[1]) sec = 0;
  else if (b < A.goff[2]) sec = 1;
  else if (b < A.goff[3]) sec = 2;
  const int64_t bis = b - A.goff[sec];

  float maxd;
  switch (sec) {
    case 0: maxd = pr_rows<4>(A, 0, bis); break;
    case 1: maxd = pr_rows<16>(A, 1, bis); break;
    case 2: maxd = pr_rows<64>(A, 2, bis); break;
    default: maxd = pr_rows<256>(A, 3, bis); break;
  }

  if (A.delta_max) {
    // block max -> one atomic per block (guide G12).
    __shared__ float wmax[kBlock / 64];
    for (int o = 32; o; o >>= 1) maxd = fmaxf(maxd, __shfl_down(maxd, o, 64));
    if ((threadIdx.x & 63) == 0) wmax[threadIdx.x >> 6] = maxd;
    __syncthreads();
    if (threadIdx.x == 0) {
      float m = wmax[0];
      for (int i = 1; i < kBlock / 64; ++i) m = fmaxf(m, wmax[i]);
      atomicMax(A.delta_max, __float_as_uint(m));
    }
  }
}

__global__ void k_pr_init(int64_t n, float r0, const float *inv, float *rank,
                          float *contrib) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    rank[i] = r0;
    contrib[i] = r0 * inv[i];
  }
}

__global__ void k_sum_f32(int64_t n, const float *x, double *out) {
  double acc = 0.0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    acc += (double)x[i];
  __shared__ double red[kBlock / 64];
  for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
  if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
  __syncthreads();
  if (threadIdx.x == 0) {
    double s = 0.0;
    for (int i = 0; i < kBlock / 64; ++i) s += red[i];
    atomicAdd(out, s);
  }
}

__global__ void k_pr_widen(int64_t n, const float *rank, const double *sum,
                           const int32_t *order, double *out) {
  // order: permuted -> original vertex id (hot-first layout); identity when
  // null. Scatter on device so the D2H stays one contiguous copy.
  const double inv = 1.0 / *sum;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[order ? order[i] : i] = (double)rank[i] * inv;
}

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

mgx_status queue_one_iteration(mgx_pagerank_run *run, bool track_delta) {
  mgx_context *ctx = run->ctx;
  mgx_graph *g = run->g;
  const int64_t V = g->n_vertices;

  PrArgs A;
  A.row_ptr = g->in_row_ptr;
  A.col = g->in_col;
  A.contrib_old = run->contrib[run->cur];
  A.rank_old = run->rank[run->cur];
  A.rank_new = run->rank[1 - run->cur];
  A.contrib_new = run->contrib[1 - run->cur];
  A.inv_outdeg = g->inv_outdeg;
  A.row_base = g->row_begin;
  A.base_term = (float)((1.0 - run->damping) / (double)V);
  A.damping = (float)run->damping;
  A.delta_max = track_delta ? run->d_delta : nullptr;
  A.partial = run->d_partial;
  {
    static const int deep = [] {
      const char *e = getenv("MGX_PR_DEEP_UNROLL");
      return e && atoi(e) ? 1 : 0;
    }();
    A.deep = deep;
  }

  if (track_delta) MGX_HIP_TRY(hipMemsetAsync(run->d_delta, 0, 4, ctx->stream));

  const int n_stripes = g->n_stripes;
  // in-CSR row count (row_end may be re-padded for the allgather shard).
  const int64_t rows = (g->row_end < V ? g->row_end : V) - g->row_begin;
  if (V > 0) {
    // Event-bracket the whole per-iteration sweep group (1 launch when
    // unstriped, n_stripes launches otherwise): feeds roofline.achieved.
    if (run->ev_used >= (int64_t)run->ev_start.size()) {
      hipEvent_t e0, e1;
      MGX_HIP_TRY(hipEventCreate(&e0));
      MGX_HIP_TRY(hipEventCreate(&e1));
      run->ev_start.push_back(e0);
      run->ev_stop.push_back(e1);
    }
    MGX_HIP_TRY(hipEventRecord(run->ev_start[run->ev_used], ctx->stream));
    for (int sIdx = 0; sIdx < n_stripes; ++sIdx) {
      const mgx_bins &bins = n_stripes == 1 ? g->bins_in : g->stripe_bins[sIdx];
      int64_t off = 0, goff = 0;
      for (int b = 0; b < 4; ++b) {
        A.n[b] = bins.count[b];
        A.off[b] = off;
        off += A.n[b];
        A.goff[b] = goff;
        A.grid[b] = bins.grid[b];
        goff += A.grid[b];
      }
      A.bin_rows = bins.rows;
      if (n_stripes == 1) {
        A.sp_lo = g->in_row_ptr;
        A.sp_hi = g->in_row_ptr + 1;
        A.mode = 0;
      } else {
        A.sp_lo = g->stripe_ptr + (size_t)sIdx * rows;
        A.sp_hi = g->stripe_ptr + (size_t)(sIdx + 1) * rows;
        A.mode = sIdx == 0 ? 1 : (sIdx == n_stripes - 1 ? 3 : 2);
      }
      if (goff > 0) {
        hipLaunchKernelGGL(k_pr_sweep, dim3((uint32_t)goff), dim3(kBlock), 0, ctx->stream,
                           A);
      }
    }
    MGX_HIP_TRY(hipEventRecord(run->ev_stop[run->ev_used], ctx->stream));
    ++run->ev_used;
    MGX_HIP_TRY(hipGetLastError());
  }

  // Distributed: every rank owns rows [row_begin,row_end); exchange the new
  // contrib slices so the next gather sees all sources (ncclAllGather over
  // xGMI). The rank vector itself is only needed at finish
  // (normalize) — gathered once there, halving the per-iteration volume.
  if (run->dist) {
    const int64_t shard = run->row_end - run->row_begin;  // equal on all ranks (padded)
    MGX_TRY(mgx_comm_allgather_f32(ctx, A.contrib_new + run->row_begin, A.contrib_new,
                                   (size_t)shard));
  }

  run->cur = 1 - run->cur;
  ++run->iterations;
  // Bounded event ring: fold timings once in a while.
  if (run->ev_used >= 1024) MGX_TRY(run->flush_timing());
  return MGX_OK;
}

}  // namespace

mgx_status mgx_pagerank_run::flush_timing() {
  if (ev_used == 0) return MGX_OK;
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  for (int64_t i = 0; i < ev_used; ++i) {
    float ms = 0.f;
    MGX_HIP_TRY(hipEventElapsedTime(&ms, ev_start[i], ev_stop[i]));
    sweep_ms_acc += ms;
  }
  launches_acc += ev_used;
  ev_used = 0;
  return MGX_OK;
}

mgx_status mgx_pagerank_queue_iterations(mgx_pagerank_run *run, int64_t n, bool track_delta) {
  for (int64_t i = 0; i < n; ++i) MGX_TRY(queue_one_iteration(run, track_delta));
  return MGX_OK;
}

mgx_status mgx_pagerank_read_delta(mgx_pagerank_run *run, float *out) {
  if (run->dist && mgx_comm_world(run->ctx) > 1) {
    // The sweep's Linf is over OWNED rows only; ranks must agree on the
    // stopping decision or the next iteration's ncclAllGather deadlocks.
    // d_delta holds the f32 bits of a non-negative float (ordered-uint
    // atomicMax), so reinterpreting as float is exact; in-place max.
    MGX_TRY(mgx_comm_allreduce_max_f32(run->ctx, (const float *)run->d_delta,
                                       (float *)run->d_delta));
  }
  uint32_t bits = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&bits, run->d_delta, 4, hipMemcpyDeviceToHost,
                             run->ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(run->ctx->stream));
  union {
    uint32_t u;
    float f;
  } cv;
  cv.u = bits;
  *out = cv.f;
  return MGX_OK;
}

mgx_status mgx_pagerank_normalize_download(mgx_pagerank_run *run, double *out_rank) {
  mgx_context *ctx = run->ctx;
  const int64_t V = run->g->n_vertices;
  if (V == 0) return MGX_OK;
  if (run->dist && run->iterations > 0) {
    // Assemble the full rank vector once (deferred from the iterations).
    const int64_t shard = run->row_end - run->row_begin;
    MGX_TRY(mgx_comm_allgather_f32(ctx, run->rank[run->cur] + run->row_begin,
                                   run->rank[run->cur], (size_t)shard));
  }
  // NormalizeRank (reference pagerank.cpp:157-162): divide by the sum.
  MGX_HIP_TRY(hipMemsetAsync(run->d_scratch, 0, sizeof(double), ctx->stream));
  hipLaunchKernelGGL(k_sum_f32, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, run->rank[run->cur], run->d_scratch);
  hipLaunchKernelGGL(k_pr_widen, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                     V, run->rank[run->cur], run->d_scratch, run->g->order,
                     run->d_scratch + 1);
  if (out_rank) {
    MGX_HIP_TRY(hipMemcpyAsync(out_rank, run->d_scratch + 1, V * sizeof(double),
                               hipMemcpyDeviceToHost, ctx->stream));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  return MGX_OK;
}

namespace {

mgx_status pagerank_start_common(mgx_context *ctx, mgx_graph *g, double damping,
                                 bool dist, int64_t row_begin, int64_t row_end,
                                 mgx_pagerank_run **out) {
  if (!ctx || !g || !(g->flags & MGX_BUILD_IN_CSR)) {
    mgx_set_error("pagerank needs a graph built with MGX_BUILD_IN_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  auto *run = new mgx_pagerank_run();
  run->ctx = ctx;
  run->g = g;
  run->damping = damping;
  run->dist = dist;
  run->row_begin = row_begin;
  run->row_end = row_end;
  const int64_t V = g->n_vertices;
  // Distributed rank/contrib arrays are padded to world*shard.
  const int64_t alloc = dist ? (row_end - row_begin) * (int64_t)mgx_comm_world(ctx) : V;
  const int64_t n = (alloc > V ? alloc : V);
  for (int i = 0; i < 2; ++i) {
    MGX_TRY(ctx->alloc_async((void **)&run->rank[i], n * sizeof(float)));
    MGX_TRY(ctx->alloc_async((void **)&run->contrib[i], n * sizeof(float)));
  }
  MGX_TRY(ctx->alloc_async((void **)&run->d_delta, sizeof(uint32_t)));
  MGX_TRY(ctx->alloc_async((void **)&run->d_scratch, (V + 1) * sizeof(double)));
  if (g->n_stripes > 1) {
    const int64_t rows = (dist ? (row_end < V ? row_end : V) : V) - row_begin;
    MGX_TRY(ctx->alloc_async((void **)&run->d_partial, rows * sizeof(double)));
  }
  if (V > 0) {
    const float r0 = (float)(1.0 / (double)V);
    hipLaunchKernelGGL(k_pr_init, dim3((uint32_t)grid_for(V)), dim3(kBlock), 0, ctx->stream,
                       V, r0, g->inv_outdeg, run->rank[0], run->contrib[0]);
    MGX_HIP_TRY(hipGetLastError());
  }
  *out = run;
  return MGX_OK;
}

void pagerank_run_free(mgx_pagerank_run *run) {
  for (int i = 0; i < 2; ++i) {
    if (run->rank[i]) (void)run->ctx->free_async(run->rank[i]);
    if (run->contrib[i]) (void)run->ctx->free_async(run->contrib[i]);
  }
  if (run->d_delta) (void)run->ctx->free_async(run->d_delta);
  if (run->d_scratch) (void)run->ctx->free_async(run->d_scratch);
  if (run->d_partial) (void)run->ctx->free_async(run->d_partial);
  for (auto e : run->ev_start) (void)hipEventDestroy(e);
  for (auto e : run->ev_stop) (void)hipEventDestroy(e);
  delete run;
}

}  // namespace

extern "C" mgx_status mgx_pagerank_start(mgx_context *ctx, mgx_graph *g, double damping,
                                         mgx_pagerank_run **out) {
  return pagerank_start_common(ctx, g, damping, false, 0, g ? g->n_vertices : 0, out);
}

extern "C" mgx_status mgx_pagerank_start_dist(mgx_context *ctx, mgx_graph *g, double damping,
                                              int64_t row_begin, int64_t row_end,
                                              mgx_pagerank_run **out) {
  if (mgx_comm_world(ctx) == 0) {
    mgx_set_error("mgx_comm_init must be called before mgx_pagerank_start_dist");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  return pagerank_start_common(ctx, g, damping, true, row_begin, row_end, out);
}

extern "C" mgx_status mgx_pagerank_iterate(mgx_pagerank_run *run, int64_t n) {
  return mgx_pagerank_queue_iterations(run, n, /*track_delta=*/false);
}

extern "C" mgx_status mgx_pagerank_iterate_eps(mgx_pagerank_run *run, int64_t max_n,
                                               double stop_epsilon, int64_t *done) {
  int64_t i = 0;
  for (; i < max_n; ++i) {
    MGX_TRY(mgx_pagerank_queue_iterations(run, 1, /*track_delta=*/true));
    float delta = 0.f;
    MGX_TRY(mgx_pagerank_read_delta(run, &delta));  // dist: max-allreduced
    if (delta <= (float)stop_epsilon) {
      ++i;
      break;
    }
  }
  if (done) *done = i;
  return MGX_OK;
}

extern "C" mgx_status mgx_pagerank_timing(mgx_pagerank_run *run, double *sweep_ms,
                                          int64_t *launches) {
  MGX_TRY(run->flush_timing());
  if (sweep_ms) *sweep_ms = run->sweep_ms_acc;
  if (launches) *launches = run->launches_acc;
  return MGX_OK;
}

extern "C" mgx_status mgx_pagerank_finish(mgx_pagerank_run *run, double *out_rank) {
  mgx_status s = mgx_pagerank_normalize_download(run, out_rank);
  if (s == MGX_OK) s = run->flush_timing();
  pagerank_run_free(run);
  return s;
}

extern "C" mgx_status mgx_pagerank(mgx_context *ctx, mgx_graph *g, int64_t max_iterations,
                                   double damping, double stop_epsilon, double *out_rank,
                                   mgx_pagerank_stats *stats) {
  mgx_pagerank_run *run = nullptr;
  MGX_TRY(mgx_pagerank_start(ctx, g, damping, &run));
  hipEvent_t it0, it1;
  MGX_HIP_TRY(hipEventCreate(&it0));
  MGX_HIP_TRY(hipEventCreate(&it1));
  MGX_HIP_TRY(hipEventRecord(it0, ctx->stream));

  mgx_status s = MGX_OK;
  if (stop_epsilon > 0.0) {
    // Reference stopping rule (pagerank.cpp:139-151): after each iteration,
    // stop when Linf(new-old) <= eps or the cap is reached.
    for (int64_t i = 0; i < max_iterations && s == MGX_OK; ++i) {
      s = mgx_pagerank_queue_iterations(run, 1, /*track_delta=*/true);
      if (s != MGX_OK) break;
      float delta = 0.f;
      s = mgx_pagerank_read_delta(run, &delta);
      if (delta <= (float)stop_epsilon) break;
    }
  } else {
    s = mgx_pagerank_queue_iterations(run, max_iterations, false);
  }

  MGX_HIP_TRY(hipEventRecord(it1, ctx->stream));

  double download_ms = 0.0;
  if (s == MGX_OK) {
    MGX_HIP_TRY(hipEventSynchronize(it1));
    const auto t0 = std::chrono::steady_clock::now();
    s = mgx_pagerank_normalize_download(run, out_rank);
    const auto t1 = std::chrono::steady_clock::now();
    download_ms = std::chrono::duration<double, std::milli>(t1 - t0).count();
  }

  if (stats && s == MGX_OK) {
    float iter_ms = 0.f;
    MGX_HIP_TRY(hipEventElapsedTime(&iter_ms, it0, it1));
    (void)run->flush_timing();
    stats->iterations = run->iterations;
    stats->iter_ms = iter_ms;
    stats->sweep_ms = run->sweep_ms_acc;
    stats->sweep_launches = run->launches_acc;
    stats->csr_build_ms = g->build_ms;
    stats->download_ms = download_ms;
  }
  MGX_HIP_TRY(hipEventDestroy(it0));
  MGX_HIP_TRY(hipEventDestroy(it1));
  pagerank_run_free(run);
  return s;
}
