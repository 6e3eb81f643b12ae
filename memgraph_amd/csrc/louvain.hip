// Louvain community detection on gfx950 — replaces the grappolo basic path
// the reference community_detection module runs with coloring=false:
//   - Jacobi sweep: parallelLouvianMethod (grappolo
//     BasicCommunitiesDetection/parallelLouvainMethod.cpp:65-290) — per
//     vertex, aggregate neighbour-community weights, pick the max-dQ
//     community with grappolo's exact rule (Utility/
//     utilityClusteringFunctions.cpp max():275-310: strict argmax, nonzero
//     ties to the smaller community id, singleton-swap protection), update
//     community degree/size deltas, modularity from the PRE-update state;
//     result is the assignment BEFORE the last sweep (pastCommAss), and the
//     empty-adjacency -1 target propagates through the triple rotation —
//     all replicated (see oracle/src/louvain_oracle.cpp for the pinned
//     sequential restatement).
//   - renumberClustersContiguously (Utility/buildNextPhase.cpp:49-78):
//     first-seen order == ascending order of each community's minimum
//     vertex index — computed with atomicMin + a stable sort.
//   - coarsening buildNextLevelGraphOpt (Utility/buildNextPhase.cpp:82-):
//     (C[i] >= C[tail]) pair aggregation by sort + reduce_by_key, self-loop
//     entry always present, mirrored lower pairs.
//   - phase loop runMultiPhaseBasic (BasicCommunitiesDetection/
//     runMultiPhaseBasic.cpp:53-146).
//
// Device layout: per-level sym CSR (u32 row_ptr, i32 col, f64 weights —
// level 0 widened from the graph's f32), community-keyed open-addressing
// hash tables (LDS per wave for rows with deg < 256, a global pool with one
// power-of-two region per hub row otherwise).
//
// Known divergence (documented, tests pin partitions on goldens +
// modularity at tolerance elsewhere): e_xx and community-degree updates use
// fp64 atomics, so summation order differs from the sequential oracle —
// sub-1e-12 modularity noise that can only matter within 1e-12 of the
// stopping threshold.

#include <cstring>

#include <rocprim/rocprim.hpp>

#include "mgx_internal.h"

namespace {

constexpr int kBlock = 256;
constexpr uint32_t kSmallRowDeg = 256;  // <: wave+LDS table; >=: block+pool
constexpr int kLdsCap = 512;            // per-wave table entries (>= 2*255)

inline bool louvain_trace() {
  // MGX_LOUVAIN_TRACE: bare stderr phase markers with NO syncs and NO
  // allocations — under AMD_SERIALIZE_KERNEL the process aborts inside the
  // faulting launch, so the last marker identifies the kernel without
  // perturbing timing or the allocation layout (the full DEBUG mode's
  // extra buffers/syncs mask the storm fault — r02 finding).
  static const bool v = [] {
    const char *e = getenv("MGX_LOUVAIN_TRACE");
    return e && atoi(e) != 0;
  }();
  return v;
}

#define MGX_LTRACE(...)                                                       \
  do {                                                                        \
    if (louvain_trace()) {                                                    \
      fprintf(stderr, "[ltrace] " __VA_ARGS__);                               \
      fprintf(stderr, "\n");                                                  \
      fflush(stderr);                                                         \
    }                                                                         \
  } while (0)

inline int louvain_diag(const char *name) {
  // diagnostic-only switches for the storm-fault bisect (r02):
  //   MGX_LOUVAIN_SKIP_SMALL / _SKIP_BIG: omit a sweep kernel (results
  //   WRONG — bisect only), MGX_LOUVAIN_SYNC_AFTER: sync+getLastError
  //   after each sweep launch and print the status.
  const char *e = getenv(name);
  return e && atoi(e) != 0;
}

inline bool louvain_debug() {
  static const bool v = [] {
    const char *e = getenv("MGX_LOUVAIN_DEBUG");
    return e && atoi(e) != 0;
  }();
  return v;
}

#define MGX_LDBG(ctx, ...)                                                    \
  do {                                                                        \
    if (louvain_debug()) {                                                    \
      hipError_t _se = hipStreamSynchronize((ctx)->stream);                   \
      hipError_t _ke = hipGetLastError();                                     \
      fprintf(stderr, "[louvain] " __VA_ARGS__);                              \
      fprintf(stderr, " sync=%s last=%s\n", hipGetErrorString(_se),          \
              hipGetErrorString(_ke));                                        \
      fflush(stderr);                                                         \
    }                                                                         \
  } while (0)

inline int64_t grid_for(int64_t work, int64_t cap = 4096) {
  int64_t g = (work + kBlock - 1) / kBlock;
  if (g < 1) g = 1;
  return g > cap ? cap : g;
}

struct Level {
  int64_t nv = 0;
  uint32_t *row_ptr = nullptr;  // [nv+1]
  int32_t *col = nullptr;       // [ne2]
  double *w = nullptr;          // [ne2]
  int64_t ne2 = 0;              // entries (each input edge twice)
  bool arena = false;           // col/w live in ctx->louv_* (not freed here)
};

// ---------- generic small kernels ----------------------------------------

__global__ void k_f32_to_f64(int64_t n, const float *in, double *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (double)in[i];
}

// sumVertexDegree (utilityClusteringFunctions.cpp:68-85), wave-per-row over
// the small list and block-per-row over the big list (a thread-per-row scan
// serialized 1M-entry hub rows: 121 ms/level measured at RMAT-24).
__global__ void k_row_wsum_init(int64_t nv, double *vdeg, double *cinfo_deg,
                                int32_t *cinfo_size) {
  for (int64_t v = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; v < nv;
       v += (int64_t)gridDim.x * blockDim.x) {
    vdeg[v] = 0.0;
    cinfo_deg[v] = 0.0;
    cinfo_size[v] = 1;
  }
}

__global__ void k_row_wsum_small(int64_t n_small, const int32_t *small_rows,
                                 const uint32_t *row_ptr, const double *w, double *vdeg,
                                 double *cinfo_deg) {
  // Thread per small row (deg < kSmallRowDeg: bounded loop, no hub tails).
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_small;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t row = small_rows[i];
    double acc = 0.0;
    for (uint32_t j = row_ptr[row]; j < row_ptr[row + 1]; ++j) acc += w[j];
    vdeg[row] = acc;
    cinfo_deg[row] = acc;
  }
}

__global__ void __launch_bounds__(kBlock) k_row_wsum_big(
    int64_t n_big, const int32_t *big_rows, const uint32_t *row_ptr, const double *w,
    double *vdeg, double *cinfo_deg) {
  __shared__ double red[kBlock / 64];
  for (int64_t bi = blockIdx.x; bi < n_big; bi += gridDim.x) {
    const int32_t row = big_rows[bi];
    double acc = 0.0;
    for (uint32_t j = row_ptr[row] + threadIdx.x; j < row_ptr[row + 1]; j += kBlock)
      acc += w[j];
    for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
    if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
    __syncthreads();
    if (threadIdx.x == 0) {
      double t = red[0] + red[1] + red[2] + red[3];
      vdeg[row] = t;
      cinfo_deg[row] = t;
    }
    __syncthreads();
  }
}

__global__ void k_sum_f64(int64_t n, const double *x, double *out) {
  double acc = 0.0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    acc += x[i];
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

__global__ void k_sum_sq_f64(int64_t n, const double *x, double *out) {
  double acc = 0.0;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    acc += x[i] * x[i];
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

__global__ void k_iota_i32(int64_t n, int32_t *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = (int32_t)i;
}

__global__ void k_fill_i32(int64_t n, int32_t v, int32_t *p) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    p[i] = v;
}

__global__ void k_apply_updates(int64_t nv, double *cinfo_deg, int32_t *cinfo_size,
                                double *cupd_deg, int32_t *cupd_size) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    cinfo_deg[i] += cupd_deg[i];
    cinfo_size[i] += cupd_size[i];
    cupd_deg[i] = 0.0;
    cupd_size[i] = 0;
  }
}

// ---------- the sweep -----------------------------------------------------

struct SweepArgs {
  const uint32_t *row_ptr;
  const int32_t *col;
  const double *w;
  const double *vdeg;
  const double *cinfo_deg;
  const int32_t *cinfo_size;
  const int32_t *curr;
  int32_t *target;
  double *cupd_deg;
  int32_t *cupd_size;
  double *e_xx;  // += Counter[0] per vertex (clusterWeightInternal sum)
  double constant;
  // work lists
  const int32_t *small_rows;
  int64_t n_small;
  const int32_t *big_rows;
  int64_t n_big;
  // global hash pool for big rows
  int32_t *pool_keys;
  double *pool_vals;
  const uint64_t *pool_off;  // [n_big+1]
  uint64_t pool_total;
  int inkernel_clear;        // experiment flag (MGX_LOUVAIN_INKERNEL_CLEAR)
  int64_t nv;                // level vertex count (community-id bound)
  int diag_mask;             // storm-bisect arms (MGX_LOUVAIN_DIAG bitmask):
                             // 1: no cupd atomics, 2: no e_xx add,
                             // 4: targets forced to stay (sc)
  // Always-on guard accounting (4 u32 slots, device):
  //   [0] region-invariant violation bits (big-row pool geometry)
  //   [1] out-of-range max_index count (the suspected OOB vector of the
  //       flaky storm-regime fault — VERDICT r01 weak #2 / ADVICE high)
  //   [2] first offending row   (0xFFFFFFFF = none)
  //   [3] first offending cid   (0xFFFFFFFF = none)
  uint32_t *guard;
};

__device__ inline int32_t guard_max_index(const SweepArgs &A, int32_t max_index,
                                          int32_t sc, int32_t row) {
  if (max_index < 0 || max_index >= A.nv) {
    atomicAdd(&A.guard[1], 1u);
    atomicCAS(&A.guard[2], 0xFFFFFFFFu, (uint32_t)row);
    atomicCAS(&A.guard[3], 0xFFFFFFFFu, (uint32_t)max_index);
    return sc;  // stay put instead of indexing cinfo/cupd out of bounds
  }
  return max_index;
}

__device__ inline void wave_lds_fence() {
  // Wave-level LDS completion + compiler ordering: the per-wave hash table
  // region is touched only by this wave's 64 lanes, so draining lgkmcnt
  // after the write phase makes every lane's LDS writes visible to every
  // lane's subsequent reads (no cross-wave traffic => no s_barrier needed).
  asm volatile("s_waitcnt lgkmcnt(0)" ::: "memory");
}

__device__ inline double dq_gain(double eiy, double eix, double vdeg_i, double ay,
                                 double ax, double constant) {
  return 2.0 * (eiy - eix) - 2.0 * vdeg_i * (ay - ax) * constant;
}

// Deterministic argmax with grappolo's rule: strictly positive best gain,
// ties to the smallest community id; otherwise stay.
struct Best {
  double gain;
  int32_t cid;
};

// One wave processes one small row using its LDS table region.
__global__ void __launch_bounds__(kBlock) k_sweep_small(SweepArgs A) {
  __shared__ int32_t keys[kBlock / 64][kLdsCap];
  __shared__ double vals[kBlock / 64][kLdsCap];
  const int wave = threadIdx.x >> 6;
  const int lane = threadIdx.x & 63;
  const int64_t waves_per_grid = (int64_t)gridDim.x * (kBlock / 64);
  for (int64_t wi = (int64_t)blockIdx.x * (kBlock / 64) + wave; wi < A.n_small;
       wi += waves_per_grid) {
    const int32_t row = A.small_rows[wi];
    const uint32_t adj1 = A.row_ptr[row], adj2 = A.row_ptr[row + 1];
    const int32_t sc = A.curr[row];
    // clear table
    for (int s = lane; s < kLdsCap; s += 64) {
      keys[wave][s] = -1;
      vals[wave][s] = 0.0;
    }
    wave_lds_fence();
    // seed own community (buildLocalMapCounter seeds Counter[0]=0)
    if (lane == 0) {
      uint32_t h = ((uint32_t)sc * 2654435761u) & (kLdsCap - 1);
      keys[wave][h] = sc;  // first insert, empty table: no probe needed
    }
    wave_lds_fence();
    double self_loop = 0.0;
    for (uint32_t j = adj1 + lane; j < adj2; j += 64) {
      const int32_t nb = A.col[j];
      const double wj = A.w[j];
      if (nb == row) self_loop += wj;
      const int32_t cid = A.curr[nb];
      uint32_t h = ((uint32_t)cid * 2654435761u) & (kLdsCap - 1);
      while (true) {
        int32_t prev = atomicCAS(&keys[wave][h], -1, cid);
        if (prev == -1 || prev == cid) break;
        h = (h + 1) & (kLdsCap - 1);
      }
      atomicAdd(&vals[wave][h], wj);
    }
    for (int o = 32; o; o >>= 1) self_loop += __shfl_down(self_loop, o, 64);
    self_loop = __shfl(self_loop, 0, 64);
    wave_lds_fence();
    // lookup own bucket
    double own = 0.0;
    {
      uint32_t h = ((uint32_t)sc * 2654435761u) & (kLdsCap - 1);
      int probes = 0;
      while (keys[wave][h] != sc && ++probes <= kLdsCap) h = (h + 1) & (kLdsCap - 1);
      if (probes > kLdsCap) {
        atomicOr(&A.guard[0], 16u);  // own-community entry vanished from LDS
      } else {
        own = vals[wave][h];
      }
    }
    const double eix = own - self_loop;
    const double ax = A.cinfo_deg[sc] - A.vdeg[row];
    Best best{0.0, sc};
    for (int s = lane; s < kLdsCap; s += 64) {
      const int32_t cid = keys[wave][s];
      if (cid >= 0 && cid != sc) {
        const double gain =
            dq_gain(vals[wave][s], eix, A.vdeg[row], A.cinfo_deg[cid], ax, A.constant);
        if (gain > best.gain || (gain == best.gain && gain != 0.0 && cid < best.cid))
          best = Best{gain, cid};
      }
    }
    for (int o = 32; o; o >>= 1) {
      Best other{__shfl_down(best.gain, o, 64), __shfl_down(best.cid, o, 64)};
      if (other.gain > best.gain ||
          (other.gain == best.gain && other.gain != 0.0 && other.cid < best.cid))
        best = other;
    }
    if (lane == 0) {
      int32_t max_index = guard_max_index(A, (best.gain > 0.0) ? best.cid : sc, sc, row);
      // swap protection (max(), utilityClusteringFunctions.cpp:305-307)
      if (A.cinfo_size[max_index] == 1 && A.cinfo_size[sc] == 1 && max_index > sc)
        max_index = sc;
      if (A.diag_mask & 4) max_index = sc;
      A.target[row] = max_index;
      if (max_index != sc && !(A.diag_mask & 1)) {
        atomicAdd(&A.cupd_deg[max_index], A.vdeg[row]);
        atomicAdd(&A.cupd_size[max_index], 1);
        atomicAdd(&A.cupd_deg[sc], -A.vdeg[row]);
        atomicAdd(&A.cupd_size[sc], -1);
      }
      if (!(A.diag_mask & 2)) atomicAdd(A.e_xx, own);
    }
    wave_lds_fence();
  }
}

// One 256-thread block per big row, table region in the global pool.
__global__ void __launch_bounds__(kBlock) k_sweep_big(SweepArgs A) {
  __shared__ double s_red[kBlock / 64];
  __shared__ int32_t s_cid[kBlock / 64];
  for (int64_t bi = blockIdx.x; bi < A.n_big; bi += gridDim.x) {
    const int32_t row = A.big_rows[bi];
    const uint32_t adj1 = A.row_ptr[row], adj2 = A.row_ptr[row + 1];
    const int32_t sc = A.curr[row];
    const uint64_t t0 = A.pool_off[bi], t1 = A.pool_off[bi + 1];
    const uint32_t cap = (uint32_t)(t1 - t0);  // power of two
    int32_t *keys = A.pool_keys + t0;
    double *vals = A.pool_vals + t0;
    // Region invariants (diagnosing the in-kernel-clear fault): offsets
    // monotone, power-of-two capacity, region within the pool.
    if (threadIdx.x == 0) {
      if (t1 <= t0 || t1 > A.pool_total || (cap & (cap - 1)) != 0 ||
          cap < 2 * (adj2 - adj1))
        atomicOr(&A.guard[0], 1u);
    }
    if (A.inkernel_clear) {
      for (uint32_t sIdx = threadIdx.x; sIdx < cap; sIdx += kBlock) {
        keys[sIdx] = -1;
        vals[sIdx] = 0.0;
      }
      __syncthreads();
    }
    // Otherwise the pool is cleared before the launch (the in-kernel
    // variant was implicated in memory faults at RMAT-24; see git history).
    if (threadIdx.x == 0) {
      uint32_t h = ((uint32_t)sc * 2654435761u) & (cap - 1);
      keys[h] = sc;
    }
    __syncthreads();
    double self_loop = 0.0;
    for (uint32_t j = adj1 + threadIdx.x; j < adj2; j += kBlock) {
      const int32_t nb = A.col[j];
      const double wj = A.w[j];
      if (nb == row) self_loop += wj;
      const int32_t cid = A.curr[nb];
      uint32_t h = ((uint32_t)cid * 2654435761u) & (cap - 1);
      while (true) {
        int32_t prev = atomicCAS(&keys[h], -1, cid);
        if (prev == -1 || prev == cid) break;
        h = (h + 1) & (cap - 1);
      }
      atomicAdd(&vals[h], wj);
    }
    // block sum of self_loop
    for (int o = 32; o; o >>= 1) self_loop += __shfl_down(self_loop, o, 64);
    if ((threadIdx.x & 63) == 0) s_red[threadIdx.x >> 6] = self_loop;
    __syncthreads();
    double sl = s_red[0] + s_red[1] + s_red[2] + s_red[3];
    __syncthreads();
    double own = 0.0;
    {
      uint32_t h = ((uint32_t)sc * 2654435761u) & (cap - 1);
      uint32_t probes = 0;
      while (keys[h] != sc && ++probes <= cap) h = (h + 1) & (cap - 1);
      if (probes > cap) {
        atomicOr(&A.guard[0], 32u);  // own-community entry vanished from pool
      } else {
        own = vals[h];
      }
    }
    const double eix = own - sl;
    const double ax = A.cinfo_deg[sc] - A.vdeg[row];
    Best best{0.0, sc};
    for (uint32_t s = threadIdx.x; s < cap; s += kBlock) {
      const int32_t cid = keys[s];
      if (cid >= 0 && cid != sc) {
        const double gain =
            dq_gain(vals[s], eix, A.vdeg[row], A.cinfo_deg[cid], ax, A.constant);
        if (gain > best.gain || (gain == best.gain && gain != 0.0 && cid < best.cid))
          best = Best{gain, cid};
      }
    }
    for (int o = 32; o; o >>= 1) {
      Best other{__shfl_down(best.gain, o, 64), __shfl_down(best.cid, o, 64)};
      if (other.gain > best.gain ||
          (other.gain == best.gain && other.gain != 0.0 && other.cid < best.cid))
        best = other;
    }
    if ((threadIdx.x & 63) == 0) {
      s_red[threadIdx.x >> 6] = best.gain;
      s_cid[threadIdx.x >> 6] = best.cid;
    }
    __syncthreads();
    if (threadIdx.x == 0) {
      for (int i = 1; i < kBlock / 64; ++i) {
        Best other{s_red[i], s_cid[i]};
        if (other.gain > best.gain ||
            (other.gain == best.gain && other.gain != 0.0 && other.cid < best.cid))
          best = other;
      }
      int32_t max_index = guard_max_index(A, (best.gain > 0.0) ? best.cid : sc, sc, row);
      if (A.cinfo_size[max_index] == 1 && A.cinfo_size[sc] == 1 && max_index > sc)
        max_index = sc;
      A.target[row] = max_index;
      if (max_index != sc) {
        atomicAdd(&A.cupd_deg[max_index], A.vdeg[row]);
        atomicAdd(&A.cupd_size[max_index], 1);
        atomicAdd(&A.cupd_deg[sc], -A.vdeg[row]);
        atomicAdd(&A.cupd_size[sc], -1);
      }
      atomicAdd(A.e_xx, own);
    }
    __syncthreads();
  }
}

// Rows with no edges: target = -1 (parallelLouvainMethod.cpp:209-211).
__global__ void k_sweep_empty(int64_t nv, const uint32_t *row_ptr, int32_t *target) {
  for (int64_t v = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; v < nv;
       v += (int64_t)gridDim.x * blockDim.x) {
    if (row_ptr[v] == row_ptr[v + 1]) target[v] = -1;
  }
}

// ---------- renumber ------------------------------------------------------

__global__ void k_rep_min(int64_t nv, const int32_t *C, int32_t *rep) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (C[i] >= 0) atomicMin(&rep[C[i]], (int32_t)i);
  }
}

__global__ void k_newid_from_sorted(int64_t n_active, const int32_t *sorted_c,
                                    int32_t *newid) {
  for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < n_active;
       k += (int64_t)gridDim.x * blockDim.x)
    newid[sorted_c[k]] = (int32_t)k;
}

__global__ void k_remap(int64_t nv, int32_t *C, const int32_t *newid) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (C[i] >= 0) C[i] = newid[C[i]];
  }
}

// ---------- coarsening ----------------------------------------------------

__global__ void k_pair_keys(int64_t ne2, const int32_t *col, const uint32_t *row_ptr,
                            int64_t nv, const int32_t *C, const double *w,
                            const int32_t *row_of_entry, uint64_t *keys, double *vals,
                            uint64_t sentinel) {
  for (int64_t j = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; j < ne2;
       j += (int64_t)gridDim.x * blockDim.x) {
    const int32_t i = row_of_entry[j];
    const int32_t ci = C[i];
    const int32_t ct = C[col[j]];
    if (ci >= ct) {
      keys[j] = ((uint64_t)(uint32_t)ci << 32) | (uint32_t)ct;
      vals[j] = w[j];
    } else {
      keys[j] = sentinel;  // sorts last, dropped
      vals[j] = 0.0;
    }
  }
}

__global__ void k_row_of_entry_small(int64_t n_small, const int32_t *rows,
                                     const uint32_t *row_ptr, int32_t *row_of_entry) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_small;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t row = rows[i];
    for (uint32_t j = row_ptr[row]; j < row_ptr[row + 1]; ++j) row_of_entry[j] = row;
  }
}

__global__ void k_row_of_entry_big(int64_t n_big, const int32_t *rows,
                                   const uint32_t *row_ptr, int32_t *row_of_entry) {
  for (int64_t bi = blockIdx.x; bi < n_big; bi += gridDim.x) {
    const int32_t row = rows[bi];
    for (uint32_t j = row_ptr[row] + threadIdx.x; j < row_ptr[row + 1]; j += kBlock)
      row_of_entry[j] = row;
  }
}

// diagnostic: count non-sentinel keys with components outside [0, ncl)
__global__ void k_validate_keys(int64_t n, const uint64_t *keys, int64_t ncl,
                                uint64_t sentinel, unsigned long long *bad) {
  unsigned long long acc = 0;
  for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < n;
       k += (int64_t)gridDim.x * blockDim.x) {
    const uint64_t key = keys[k];
    if (key == sentinel) continue;
    const int64_t ci = (int64_t)(key >> 32);
    const int64_t ct = (int64_t)(uint32_t)key;
    if (ci < 0 || ci >= ncl || ct < 0 || ct >= ncl) ++acc;
  }
  if (acc) atomicAdd(bad, acc);
}

// count CSR rows for unique pairs (ci>=ct): row ci +1; if ct<ci also row ct +1.
__global__ void k_pair_counts(int64_t n_pairs, const uint64_t *keys, int64_t ncl,
                              uint32_t *counts, uint32_t *self_present,
                              unsigned long long *oob) {
  for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < n_pairs;
       k += (int64_t)gridDim.x * blockDim.x) {
    const int32_t ci = (int32_t)(keys[k] >> 32);
    const int32_t ct = (int32_t)(uint32_t)keys[k];
    // Bounds: a corrupt pair id must be accounted, never become an OOB
    // atomicAdd (the r02 storm fault vector — counts[1e9] faults or
    // silently tramples a neighbouring allocation).
    if (ci < 0 || ci >= ncl || ct < 0 || ct > ci) {
      atomicAdd(oob, 1ull);
      continue;
    }
    atomicAdd(&counts[ci], 1u);
    if (ct < ci) atomicAdd(&counts[ct], 1u);
    else self_present[ci] = 1u;  // ct == ci
  }
}

__global__ void k_add_missing_self(int64_t ncl, const uint32_t *self_present,
                                   uint32_t *counts) {
  for (int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; c < ncl;
       c += (int64_t)gridDim.x * blockDim.x) {
    if (!self_present[c]) atomicAdd(&counts[c], 1u);
  }
}

__global__ void k_pair_scatter(int64_t n_pairs, const uint64_t *keys, const double *vals,
                               int64_t ncl, uint32_t total, uint32_t *cursor, int32_t *col,
                               double *w) {
  for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < n_pairs;
       k += (int64_t)gridDim.x * blockDim.x) {
    const int32_t ci = (int32_t)(keys[k] >> 32);
    const int32_t ct = (int32_t)(uint32_t)keys[k];
    if (ci < 0 || ci >= ncl || ct < 0 || ct > ci) continue;  // accounted in counts
    uint32_t p = atomicAdd(&cursor[ci], 1u);
    if (p < total) {
      col[p] = ct;
      w[p] = vals[k];
    }
    if (ct < ci) {
      p = atomicAdd(&cursor[ct], 1u);
      if (p < total) {
        col[p] = ci;
        w[p] = vals[k];
      }
    }
  }
}

__global__ void k_self_scatter(int64_t ncl, const uint32_t *self_present, uint32_t total,
                               uint32_t *cursor, int32_t *col, double *w) {
  for (int64_t c = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; c < ncl;
       c += (int64_t)gridDim.x * blockDim.x) {
    if (!self_present[c]) {
      uint32_t p = atomicAdd(&cursor[c], 1u);
      if (p < total) {
        col[p] = (int32_t)c;
        w[p] = 0.0;  // zero-weight self loop (buildNextPhase.cpp cluPtrIn init)
      }
    }
  }
}

// big-row pool capacities (power of two >= 2*deg)
__global__ void k_big_caps(int64_t n_big, const int32_t *big_rows, const uint32_t *row_ptr,
                           uint64_t *caps) {
  for (int64_t k = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; k < n_big;
       k += (int64_t)gridDim.x * blockDim.x) {
    const int32_t row = big_rows[k];
    uint64_t need = 2ull * (row_ptr[row + 1] - row_ptr[row]) + 2;
    uint64_t cap = 1;
    while (cap < need) cap <<= 1;
    caps[k] = cap;
  }
}

__global__ void k_classify_rows(int64_t nv, const uint32_t *row_ptr, uint32_t *n_small,
                                uint32_t *n_big) {
  __shared__ uint32_t local[2];
  if (threadIdx.x < 2) local[threadIdx.x] = 0;
  __syncthreads();
  for (int64_t v = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; v < nv;
       v += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t deg = row_ptr[v + 1] - row_ptr[v];
    if (deg == 0) continue;
    atomicAdd(&local[deg < kSmallRowDeg ? 0 : 1], 1u);
  }
  __syncthreads();
  if (threadIdx.x == 0 && local[0]) atomicAdd(n_small, local[0]);
  if (threadIdx.x == 1 && local[1]) atomicAdd(n_big, local[1]);
}

__global__ void k_fill_rows(int64_t nv, const uint32_t *row_ptr, int32_t *small_rows,
                            int32_t *big_rows, uint32_t *c_small, uint32_t *c_big) {
  // Order within the lists is nondeterministic (atomics) but irrelevant:
  // the sweep is a Jacobi update over a fixed snapshot. Per-block LDS
  // staging keeps the two global counters off the hot path.
  __shared__ int32_t stage_s[kBlock], stage_b[kBlock];
  __shared__ uint32_t ns, nb, base_s, base_b;
  for (int64_t start = (int64_t)blockIdx.x * kBlock; start < nv;
       start += (int64_t)gridDim.x * kBlock) {
    if (threadIdx.x == 0) ns = nb = 0;
    __syncthreads();
    const int64_t v = start + threadIdx.x;
    if (v < nv) {
      const uint32_t deg = row_ptr[v + 1] - row_ptr[v];
      if (deg != 0) {
        if (deg < kSmallRowDeg) stage_s[atomicAdd(&ns, 1u)] = (int32_t)v;
        else stage_b[atomicAdd(&nb, 1u)] = (int32_t)v;
      }
    }
    __syncthreads();
    if (threadIdx.x == 0 && ns) base_s = atomicAdd(c_small, ns);
    if (threadIdx.x == 1 && nb) base_b = atomicAdd(c_big, nb);
    __syncthreads();
    if (threadIdx.x < ns) small_rows[base_s + threadIdx.x] = stage_s[threadIdx.x];
    if (threadIdx.x < nb) big_rows[base_b + threadIdx.x] = stage_b[threadIdx.x];
    __syncthreads();
  }
}

__global__ void k_check_lists(int64_t nv, const uint32_t *row_ptr, const int32_t *small_rows,
                              int64_t n_small, const int32_t *big_rows, int64_t n_big,
                              uint32_t *mark, uint32_t *err) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n_small + n_big;
       i += (int64_t)gridDim.x * blockDim.x) {
    const int32_t row = i < n_small ? small_rows[i] : big_rows[i - n_small];
    if (row < 0 || row >= nv) {
      atomicOr(err, 1u);
      continue;
    }
    atomicAdd(&mark[row], 1u);
    const uint32_t deg = row_ptr[row + 1] - row_ptr[row];
    if (i < n_small && !(deg > 0 && deg < kSmallRowDeg)) atomicOr(err, 2u);
    if (i >= n_small && deg < kSmallRowDeg) atomicOr(err, 4u);
  }
}

__global__ void k_check_marks(int64_t nv, const uint32_t *row_ptr, const uint32_t *mark,
                              uint32_t *err) {
  for (int64_t v = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; v < nv;
       v += (int64_t)gridDim.x * blockDim.x) {
    const uint32_t deg = row_ptr[v + 1] - row_ptr[v];
    const uint32_t expect = deg > 0 ? 1u : 0u;
    if (mark[v] != expect) atomicOr(err, 8u);
  }
}

// ---------- host orchestration -------------------------------------------

struct DevBuf {
  // Allocation from the context's caching free list (mgx_internal.h): the
  // per-level multi-GB hipMalloc/hipFree pairs cost ~2 s of host time per
  // Louvain call at RMAT-24 (measured); the cache pays it once per process.
  mgx_context *ctx = nullptr;
  void *p = nullptr;
  ~DevBuf() {
    if (p) (void)ctx->free_async(p);
  }
  hipError_t alloc(mgx_context *c, size_t bytes) {
    ctx = c;
    return c->alloc_async(&p, bytes) == MGX_OK ? hipSuccess : hipErrorOutOfMemory;
  }
  template <typename T>
  T *as() {
    return (T *)p;
  }
};

mgx_status read_scalar_f64(mgx_context *ctx, const double *d, double *out) {
  MGX_HIP_TRY(hipMemcpyAsync(out, d, sizeof(double), hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  return MGX_OK;
}

// Runs parallelLouvianMethod on one level. C (device i32[nv]) receives
// pastCommAss. Returns modularity via *out_mod.
mgx_status louvain_level(mgx_context *ctx, const Level &L, double lower, double thresh,
                         int32_t *C, double *out_mod, int64_t *iters_out) {
  const int64_t nv = L.nv;
  DevBuf vdeg, cinfo_deg, cinfo_size, cupd_deg, cupd_size, past, curr, target, scalars;
  MGX_HIP_TRY(vdeg.alloc(ctx, nv * 8));
  MGX_HIP_TRY(cinfo_deg.alloc(ctx, nv * 8));
  MGX_HIP_TRY(cinfo_size.alloc(ctx, nv * 4));
  MGX_HIP_TRY(cupd_deg.alloc(ctx, nv * 8));
  MGX_HIP_TRY(cupd_size.alloc(ctx, nv * 4));
  MGX_HIP_TRY(past.alloc(ctx, nv * 4));
  MGX_HIP_TRY(curr.alloc(ctx, nv * 4));
  MGX_HIP_TRY(target.alloc(ctx, nv * 4));
  MGX_HIP_TRY(scalars.alloc(ctx, 3 * 8));  // [e_xx, a2_x, total_w]

  MGX_HIP_TRY(hipMemsetAsync(scalars.as<double>() + 2, 0, 8, ctx->stream));
  hipLaunchKernelGGL(k_iota_i32, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, past.as<int32_t>());
  hipLaunchKernelGGL(k_iota_i32, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, curr.as<int32_t>());
  MGX_HIP_TRY(hipMemsetAsync(cupd_deg.p, 0, nv * 8, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(cupd_size.p, 0, nv * 4, ctx->stream));

  // Row classification + big-row pool.
  DevBuf counters, small_rows, big_rows, pool_off, pool_keys, pool_vals;
  MGX_HIP_TRY(counters.alloc(ctx, 4 * 4));
  MGX_HIP_TRY(hipMemsetAsync(counters.p, 0, 16, ctx->stream));
  hipLaunchKernelGGL(k_classify_rows, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                     ctx->stream, nv, L.row_ptr, counters.as<uint32_t>(),
                     counters.as<uint32_t>() + 1);
  uint32_t h_counts[2] = {0, 0};
  MGX_HIP_TRY(hipMemcpyAsync(h_counts, counters.p, 8, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  const int64_t n_small = h_counts[0], n_big = h_counts[1];
  MGX_HIP_TRY(small_rows.alloc(ctx, n_small * 4));
  MGX_HIP_TRY(big_rows.alloc(ctx, n_big * 4));
  MGX_HIP_TRY(hipMemsetAsync(counters.p, 0, 16, ctx->stream));
  hipLaunchKernelGGL(k_fill_rows, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                     ctx->stream, nv, L.row_ptr, small_rows.as<int32_t>(),
                     big_rows.as<int32_t>(), counters.as<uint32_t>(),
                     counters.as<uint32_t>() + 1);
  if (louvain_debug()) {
    DevBuf mark, errbuf;
    MGX_HIP_TRY(mark.alloc(ctx, nv * 4));
    MGX_HIP_TRY(errbuf.alloc(ctx, 4));
    MGX_HIP_TRY(hipMemsetAsync(mark.p, 0, nv * 4, ctx->stream));
    MGX_HIP_TRY(hipMemsetAsync(errbuf.p, 0, 4, ctx->stream));
    hipLaunchKernelGGL(k_check_lists, dim3((uint32_t)grid_for(n_small + n_big)),
                       dim3(kBlock), 0, ctx->stream, nv, L.row_ptr,
                       small_rows.as<int32_t>(), n_small, big_rows.as<int32_t>(), n_big,
                       mark.as<uint32_t>(), errbuf.as<uint32_t>());
    hipLaunchKernelGGL(k_check_marks, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                       ctx->stream, nv, L.row_ptr, mark.as<uint32_t>(),
                       errbuf.as<uint32_t>());
    uint32_t e = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&e, errbuf.p, 4, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    fprintf(stderr, "[louvain] LIST CHECK nv=%lld err=%u\n", (long long)nv, e);
    fflush(stderr);
  }
  // Binned row weight sums (sumVertexDegree) over the just-built lists.
  MGX_LTRACE("setup k_row_wsum_init nv=%lld", (long long)nv);
  hipLaunchKernelGGL(k_row_wsum_init, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                     ctx->stream, nv, vdeg.as<double>(), cinfo_deg.as<double>(),
                     cinfo_size.as<int32_t>());
  if (n_small > 0)
    hipLaunchKernelGGL(k_row_wsum_small, dim3((uint32_t)grid_for(n_small)), dim3(kBlock),
                       0, ctx->stream, n_small, small_rows.as<int32_t>(), L.row_ptr, L.w,
                       vdeg.as<double>(), cinfo_deg.as<double>());
  if (n_big > 0)
    hipLaunchKernelGGL(k_row_wsum_big, dim3((uint32_t)(n_big < 4096 ? n_big : 4096)),
                       dim3(kBlock), 0, ctx->stream, n_big, big_rows.as<int32_t>(),
                       L.row_ptr, L.w, vdeg.as<double>(), cinfo_deg.as<double>());
  hipLaunchKernelGGL(k_sum_f64, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, vdeg.as<double>(), scalars.as<double>() + 2);
  double total_w = 0.0;
  MGX_TRY(read_scalar_f64(ctx, scalars.as<double>() + 2, &total_w));
  const double constant = 1.0 / total_w;  // calConstantForSecondTerm

  MGX_LDBG(ctx, "level nv=%lld n_small=%lld n_big=%lld wsum done", (long long)nv,
           (long long)n_small, (long long)n_big);
  uint64_t pool_total = 0;
  if (n_big > 0) {
    DevBuf caps;
    MGX_HIP_TRY(caps.alloc(ctx, n_big * 8));
    hipLaunchKernelGGL(k_big_caps, dim3((uint32_t)grid_for(n_big)), dim3(kBlock), 0,
                       ctx->stream, n_big, big_rows.as<int32_t>(), L.row_ptr,
                       caps.as<uint64_t>());
    MGX_HIP_TRY(pool_off.alloc(ctx, (n_big + 1) * 8));
    size_t tmp_bytes = 0;
    auto err = rocprim::exclusive_scan(nullptr, tmp_bytes, caps.as<uint64_t>(),
                                       pool_off.as<uint64_t>(), (uint64_t)0, n_big,
                                       rocprim::plus<uint64_t>(), ctx->stream);
    if (err != hipSuccess) return MGX_ERR_HIP;
    void *tmp = nullptr;
    MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
    err = rocprim::exclusive_scan(tmp, tmp_bytes, caps.as<uint64_t>(),
                                  pool_off.as<uint64_t>(), (uint64_t)0, n_big,
                                  rocprim::plus<uint64_t>(), ctx->stream);
    if (err != hipSuccess) return MGX_ERR_HIP;
    uint64_t last_off = 0, last_cap = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&last_off, pool_off.as<uint64_t>() + n_big - 1, 8,
                               hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipMemcpyAsync(&last_cap, caps.as<uint64_t>() + n_big - 1, 8,
                               hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    pool_total = last_off + last_cap;
    MGX_HIP_TRY(hipMemcpyAsync(pool_off.as<uint64_t>() + n_big, &pool_total, 8,
                               hipMemcpyHostToDevice, ctx->stream));
    MGX_HIP_TRY(pool_keys.alloc(ctx, pool_total * 4));
    MGX_HIP_TRY(pool_vals.alloc(ctx, pool_total * 8));
  }

  MGX_LDBG(ctx, "pool_total=%llu", (unsigned long long)pool_total);
  MGX_LTRACE("setup done pool_total=%llu n_small=%lld n_big=%lld",
             (unsigned long long)pool_total, (long long)n_small, (long long)n_big);
  SweepArgs A;
  A.row_ptr = L.row_ptr;
  A.col = L.col;
  A.w = L.w;
  A.vdeg = vdeg.as<double>();
  A.cinfo_deg = cinfo_deg.as<double>();
  A.cinfo_size = cinfo_size.as<int32_t>();
  A.cupd_deg = cupd_deg.as<double>();
  A.cupd_size = cupd_size.as<int32_t>();
  A.e_xx = scalars.as<double>();
  A.constant = constant;
  A.small_rows = small_rows.as<int32_t>();
  A.n_small = n_small;
  A.big_rows = big_rows.as<int32_t>();
  A.n_big = n_big;
  A.pool_keys = pool_keys.as<int32_t>();
  A.pool_vals = pool_vals.as<double>();
  A.pool_off = pool_off.as<uint64_t>();
  A.pool_total = pool_total;
  A.inkernel_clear = 0;
  A.nv = nv;
  {
    const char *e = getenv("MGX_LOUVAIN_DIAG");
    A.diag_mask = e ? atoi(e) : 0;
  }
  DevBuf guard;
  {
    const char *e = getenv("MGX_LOUVAIN_INKERNEL_CLEAR");
    if (e && atoi(e)) A.inkernel_clear = 1;
  }
  // Guard accounting is ALWAYS on (cost: one 16-B buffer + one lane-0 branch
  // per row — unmeasurable next to the sweep's hash traffic).
  MGX_HIP_TRY(guard.alloc(ctx, 16));
  {
    static const uint32_t init[4] = {0u, 0u, 0xFFFFFFFFu, 0xFFFFFFFFu};
    MGX_HIP_TRY(hipMemcpyAsync(guard.p, init, 16, hipMemcpyHostToDevice, ctx->stream));
  }
  A.guard = guard.as<uint32_t>();

  double prev_mod = -1.0, curr_mod = -1.0;
  int64_t iters = 0;
  int32_t *p_past = past.as<int32_t>(), *p_curr = curr.as<int32_t>(),
          *p_target = target.as<int32_t>();
  while (true) {
    ++iters;
    MGX_HIP_TRY(hipMemsetAsync(scalars.p, 0, 16, ctx->stream));  // e_xx, a2_x
    A.curr = p_curr;
    A.target = p_target;
    MGX_LDBG(ctx, "iter=%lld pre-sweep", (long long)iters);
    MGX_LTRACE("it=%lld k_sweep_empty", (long long)iters);
    hipLaunchKernelGGL(k_sweep_empty, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                       ctx->stream, nv, L.row_ptr, p_target);
    MGX_LDBG(ctx, "iter=%lld after-empty", (long long)iters);
    if (n_small > 0) {
      const int64_t waves_needed = n_small;
      const int64_t blocks = grid_for(waves_needed * 64, 4096);
      MGX_LTRACE("it=%lld k_sweep_small", (long long)iters);
      static const int skip_small = louvain_diag("MGX_LOUVAIN_SKIP_SMALL");
      if (!skip_small)
        hipLaunchKernelGGL(k_sweep_small, dim3((uint32_t)blocks), dim3(kBlock), 0,
                           ctx->stream, A);
      static const int sync_after = louvain_diag("MGX_LOUVAIN_SYNC_AFTER");
      if (sync_after) {
        hipError_t se = hipStreamSynchronize(ctx->stream);
        hipError_t le = hipGetLastError();
        if (se != hipSuccess || le != hipSuccess)
          MGX_LTRACE("it=%lld SMALL sync=%s last=%s", (long long)iters,
                     hipGetErrorString(se), hipGetErrorString(le));
      }
      MGX_LTRACE("it=%lld k_sweep_small launched", (long long)iters);
    }
    MGX_LDBG(ctx, "iter=%lld after-small", (long long)iters);
    if (n_big > 0) {
      if (!A.inkernel_clear) {
        MGX_LTRACE("it=%lld pool_clear", (long long)iters);
        hipLaunchKernelGGL(k_fill_i32, dim3((uint32_t)grid_for((int64_t)pool_total)),
                           dim3(kBlock), 0, ctx->stream, (int64_t)pool_total, -1,
                           pool_keys.as<int32_t>());
        MGX_HIP_TRY(hipMemsetAsync(pool_vals.p, 0, pool_total * 8, ctx->stream));
      }
      const int64_t blocks = n_big < 4096 ? n_big : 4096;
      MGX_LTRACE("it=%lld k_sweep_big", (long long)iters);
      static const int skip_big = louvain_diag("MGX_LOUVAIN_SKIP_BIG");
      if (!skip_big)
        hipLaunchKernelGGL(k_sweep_big, dim3((uint32_t)blocks), dim3(kBlock), 0,
                           ctx->stream, A);
      static const int sync_after_b = louvain_diag("MGX_LOUVAIN_SYNC_AFTER");
      if (sync_after_b) {
        hipError_t se = hipStreamSynchronize(ctx->stream);
        hipError_t le = hipGetLastError();
        if (se != hipSuccess || le != hipSuccess)
          MGX_LTRACE("it=%lld BIG sync=%s last=%s", (long long)iters,
                     hipGetErrorString(se), hipGetErrorString(le));
      }
      MGX_LTRACE("it=%lld k_sweep_big launched", (long long)iters);
    }
    MGX_LDBG(ctx, "iter=%lld after-big", (long long)iters);
    MGX_LDBG(ctx, "iter=%lld post-sweep", (long long)iters);
    MGX_LTRACE("it=%lld k_sum_sq", (long long)iters);
    hipLaunchKernelGGL(k_sum_sq_f64, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                       ctx->stream, nv, cinfo_deg.as<double>(), scalars.as<double>() + 1);
    double exx_a2x[2] = {0.0, 0.0};
    MGX_HIP_TRY(hipMemcpyAsync(exx_a2x, scalars.p, 16, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (louvain_trace()) {
      uint32_t ge[4] = {0, 0, 0, 0};
      MGX_HIP_TRY(hipMemcpyAsync(ge, A.guard, 16, hipMemcpyDeviceToHost, ctx->stream));
      MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
      if (ge[0] || ge[1])
        MGX_LTRACE("it=%lld GUARD bits=%u oob=%u row=%u cid=%u", (long long)iters, ge[0],
                   ge[1], ge[2], ge[3]);
    }
    curr_mod = exx_a2x[0] * constant - exx_a2x[1] * constant * constant;
    if (louvain_debug() && (iters < 10 || iters % 500 == 0)) {
      fprintf(stderr, "[louvain] iter=%lld exx=%.12g a2x=%.12g mod=%.12g prev=%.12g\n",
              (long long)iters, exx_a2x[0], exx_a2x[1], curr_mod, prev_mod);
      fflush(stderr);
    }
    if ((curr_mod - prev_mod) < thresh) break;
    prev_mod = curr_mod;
    if (prev_mod < lower) prev_mod = lower;
    MGX_LTRACE("it=%lld k_apply_updates", (long long)iters);
    hipLaunchKernelGGL(k_apply_updates, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                       ctx->stream, nv, cinfo_deg.as<double>(), cinfo_size.as<int32_t>(),
                       cupd_deg.as<double>(), cupd_size.as<int32_t>());
    // rotation (parallelLouvainMethod.cpp:268-274)
    int32_t *tmp = p_past;
    p_past = p_curr;
    p_curr = p_target;
    p_target = tmp;
    if (iters > 100000) break;  // matches runMultiPhaseBasic's totItr cap scale
  }

  {
    uint32_t ge[4] = {0, 0, 0, 0};
    MGX_HIP_TRY(hipMemcpyAsync(ge, A.guard, 16, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (ge[0] || ge[1]) {
      fprintf(stderr,
              "[louvain] GUARD FIRED region_bits=%u oob_count=%u first_row=%u "
              "first_cid=%u (nv=%lld iters=%lld) — clamped to stay-put; "
              "please report\n",
              ge[0], ge[1], ge[2], ge[3], (long long)nv, (long long)iters);
      fflush(stderr);
      const char *strict = getenv("MGX_LOUVAIN_STRICT");
      if (strict && atoi(strict)) {
        mgx_set_error("louvain device guard fired (region_bits=%u oob=%u row=%u cid=%u)",
                      ge[0], ge[1], ge[2], ge[3]);
        return MGX_ERR_HIP;
      }
    } else if (louvain_debug()) {
      fprintf(stderr, "[louvain] guard clean (nv=%lld)\n", (long long)nv);
      fflush(stderr);
    }
  }
  MGX_HIP_TRY(hipMemcpyAsync(C, p_past, nv * 4, hipMemcpyDeviceToDevice, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  // The reference's sweep loop returns prevMod — the (Lower-clamped)
  // modularity of the ADOPTED assignment, not the last computed currMod
  // (parallelLouvainMethod.cpp:307). The two differ by < thresh, which
  // flips the phase-continue decision at loose thresholds.
  *out_mod = prev_mod;
  *iters_out = iters;
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}

// renumberClustersContiguously: new id = rank of community's min vertex.
mgx_status renumber(mgx_context *ctx, int32_t *C, int64_t nv, int64_t *n_clusters) {
  DevBuf rep, keys_out, cand, cand_out, newid, nact;
  MGX_HIP_TRY(rep.alloc(ctx, nv * 4));
  hipLaunchKernelGGL(k_fill_i32, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, INT32_MAX, rep.as<int32_t>());
  hipLaunchKernelGGL(k_rep_min, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, C, rep.as<int32_t>());
  // sort (rep, c) ascending by rep; INT32_MAX reps (inactive ids) sort last.
  MGX_HIP_TRY(cand.alloc(ctx, nv * 4));
  MGX_HIP_TRY(cand_out.alloc(ctx, nv * 4));
  MGX_HIP_TRY(keys_out.alloc(ctx, nv * 4));
  hipLaunchKernelGGL(k_iota_i32, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, cand.as<int32_t>());
  size_t tmp_bytes = 0;
  auto err = rocprim::radix_sort_pairs(nullptr, tmp_bytes, (uint32_t *)rep.p,
                                       (uint32_t *)keys_out.p, cand.as<int32_t>(),
                                       cand_out.as<int32_t>(), nv, 0, 32, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::radix_sort_pairs(tmp, tmp_bytes, (uint32_t *)rep.p,
                                  (uint32_t *)keys_out.p, cand.as<int32_t>(),
                                  cand_out.as<int32_t>(), nv, 0, 32, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  // count active = reps != INT32_MAX: binary property; count via reduce on
  // host copy of the boundary — simpler: count nonzero with a kernel.
  DevBuf count;
  MGX_HIP_TRY(count.alloc(ctx, 8));
  MGX_HIP_TRY(hipMemsetAsync(count.p, 0, 8, ctx->stream));
  {
    struct K {
      static __global__ void count_active(int64_t n, const uint32_t *sorted_rep,
                                          unsigned long long *out) {
        unsigned long long acc = 0;
        for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
             i += (int64_t)gridDim.x * blockDim.x)
          if (sorted_rep[i] != 0x7FFFFFFFu) ++acc;
        __shared__ unsigned long long red[kBlock / 64];
        for (int o = 32; o; o >>= 1) acc += __shfl_down(acc, o, 64);
        if ((threadIdx.x & 63) == 0) red[threadIdx.x >> 6] = acc;
        __syncthreads();
        if (threadIdx.x == 0) {
          unsigned long long s = 0;
          for (int i = 0; i < kBlock / 64; ++i) s += red[i];
          atomicAdd(out, s);
        }
      }
    };
    hipLaunchKernelGGL(K::count_active, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0,
                       ctx->stream, nv, (uint32_t *)keys_out.p,
                       (unsigned long long *)count.p);
  }
  unsigned long long n_active = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&n_active, count.p, 8, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_LTRACE("renumber nv=%lld n_active=%llu", (long long)nv,
             (unsigned long long)n_active);
  MGX_HIP_TRY(newid.alloc(ctx, nv * 4));
  hipLaunchKernelGGL(k_newid_from_sorted, dim3((uint32_t)grid_for((int64_t)n_active)),
                     dim3(kBlock), 0, ctx->stream, (int64_t)n_active,
                     cand_out.as<int32_t>(), newid.as<int32_t>());
  hipLaunchKernelGGL(k_remap, dim3((uint32_t)grid_for(nv)), dim3(kBlock), 0, ctx->stream,
                     nv, C, newid.as<int32_t>());
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  *n_clusters = (int64_t)n_active;
  return MGX_OK;
}

// buildNextLevelGraphOpt: aggregate (C[i] >= C[tail]) pairs, always-present
// self loops, mirrored lower pairs.
mgx_status coarsen(mgx_context *ctx, const Level &in, const int32_t *C, int64_t n_clusters,
                   Level *out) {
  const int64_t ne2 = in.ne2;
  DevBuf row_of_entry, keys, vals, keys_sorted, vals_sorted, u_keys, u_vals, u_count;
  MGX_HIP_TRY(row_of_entry.alloc(ctx, ne2 * 4));
  {
    // Binned expansion of the CSR's row ids (hub rows block-parallel).
    DevBuf counters, srows, brows;
    MGX_HIP_TRY(counters.alloc(ctx, 4 * 4));
    MGX_HIP_TRY(hipMemsetAsync(counters.p, 0, 16, ctx->stream));
    hipLaunchKernelGGL(k_classify_rows, dim3((uint32_t)grid_for(in.nv)), dim3(kBlock), 0,
                       ctx->stream, in.nv, in.row_ptr, counters.as<uint32_t>(),
                       counters.as<uint32_t>() + 1);
    uint32_t hc[2] = {0, 0};
    MGX_HIP_TRY(hipMemcpyAsync(hc, counters.p, 8, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    MGX_HIP_TRY(srows.alloc(ctx, (hc[0] > 0 ? hc[0] : 1) * 4));
    MGX_HIP_TRY(brows.alloc(ctx, (hc[1] > 0 ? hc[1] : 1) * 4));
    MGX_HIP_TRY(hipMemsetAsync(counters.p, 0, 16, ctx->stream));
    hipLaunchKernelGGL(k_fill_rows, dim3((uint32_t)grid_for(in.nv)), dim3(kBlock), 0,
                       ctx->stream, in.nv, in.row_ptr, srows.as<int32_t>(),
                       brows.as<int32_t>(), counters.as<uint32_t>(),
                       counters.as<uint32_t>() + 1);
    if (hc[0] > 0)
      hipLaunchKernelGGL(k_row_of_entry_small, dim3((uint32_t)grid_for(hc[0])),
                         dim3(kBlock), 0, ctx->stream, (int64_t)hc[0], srows.as<int32_t>(),
                         in.row_ptr, row_of_entry.as<int32_t>());
    if (hc[1] > 0)
      hipLaunchKernelGGL(k_row_of_entry_big,
                         dim3((uint32_t)(hc[1] < 4096 ? hc[1] : 4096)), dim3(kBlock), 0,
                         ctx->stream, (int64_t)hc[1], brows.as<int32_t>(), in.row_ptr,
                         row_of_entry.as<int32_t>());
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  }
  MGX_HIP_TRY(keys.alloc(ctx, ne2 * 8));
  MGX_HIP_TRY(vals.alloc(ctx, ne2 * 8));
  const uint64_t sentinel = ~0ull;
  hipLaunchKernelGGL(k_pair_keys, dim3((uint32_t)grid_for(ne2)), dim3(kBlock), 0,
                     ctx->stream, ne2, in.col, in.row_ptr, in.nv, C, in.w,
                     row_of_entry.as<int32_t>(), keys.as<uint64_t>(), vals.as<double>(),
                     sentinel);
  MGX_HIP_TRY(keys_sorted.alloc(ctx, ne2 * 8));
  MGX_HIP_TRY(vals_sorted.alloc(ctx, ne2 * 8));
  DevBuf d_bad;
  MGX_HIP_TRY(d_bad.alloc(ctx, 8));
  MGX_HIP_TRY(hipMemsetAsync(d_bad.p, 0, 8, ctx->stream));
  if (louvain_trace()) {
    hipLaunchKernelGGL(k_validate_keys, dim3((uint32_t)grid_for(ne2)), dim3(kBlock), 0,
                       ctx->stream, ne2, keys.as<uint64_t>(), n_clusters, sentinel,
                       (unsigned long long *)d_bad.p);
    unsigned long long bad = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&bad, d_bad.p, 8, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    MGX_LTRACE("coarsen PAIRKEYS bad=%llu", bad);
    MGX_HIP_TRY(hipMemsetAsync(d_bad.p, 0, 8, ctx->stream));
  }
  size_t tmp_bytes = 0;
  auto err = rocprim::radix_sort_pairs(nullptr, tmp_bytes, keys.as<uint64_t>(),
                                       keys_sorted.as<uint64_t>(), vals.as<double>(),
                                       vals_sorted.as<double>(), ne2, 0, 64, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  void *tmp = nullptr;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::radix_sort_pairs(tmp, tmp_bytes, keys.as<uint64_t>(),
                                  keys_sorted.as<uint64_t>(), vals.as<double>(),
                                  vals_sorted.as<double>(), ne2, 0, 64, ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  MGX_HIP_TRY(u_keys.alloc(ctx, ne2 * 8));
  MGX_HIP_TRY(u_vals.alloc(ctx, ne2 * 8));
  MGX_HIP_TRY(u_count.alloc(ctx, 8));
  err = rocprim::reduce_by_key(nullptr, tmp_bytes, keys_sorted.as<uint64_t>(),
                               vals_sorted.as<double>(), ne2, u_keys.as<uint64_t>(),
                               u_vals.as<double>(), (unsigned int *)u_count.p,
                               rocprim::plus<double>(), rocprim::equal_to<uint64_t>(),
                               ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::reduce_by_key(tmp, tmp_bytes, keys_sorted.as<uint64_t>(),
                               vals_sorted.as<double>(), ne2, u_keys.as<uint64_t>(),
                               u_vals.as<double>(), (unsigned int *)u_count.p,
                               rocprim::plus<double>(), rocprim::equal_to<uint64_t>(),
                               ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  unsigned int n_unique = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&n_unique, u_count.p, 4, hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_LTRACE("coarsen ne2=%lld n_unique=%u", (long long)ne2, n_unique);
  if (louvain_trace() && n_unique > 0) {
    hipLaunchKernelGGL(k_validate_keys, dim3((uint32_t)grid_for((int64_t)n_unique)),
                       dim3(kBlock), 0, ctx->stream, (int64_t)n_unique,
                       u_keys.as<uint64_t>(), n_clusters, sentinel,
                       (unsigned long long *)d_bad.p);
    unsigned long long bad = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&bad, d_bad.p, 8, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    MGX_LTRACE("coarsen UKEYS bad=%llu", bad);
    MGX_HIP_TRY(hipMemsetAsync(d_bad.p, 0, 8, ctx->stream));
  }
  // drop the sentinel group if present (it sorts last)
  int64_t n_pairs = n_unique;
  if (n_pairs > 0) {
    uint64_t last_key = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&last_key, u_keys.as<uint64_t>() + n_pairs - 1, 8,
                               hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (last_key == sentinel) --n_pairs;
  }

  if (louvain_trace() && n_pairs > 0) {
    // probe: first/last unique keys + a counts sample after the passes
    uint64_t k0 = 0, k1 = 0;
    (void)hipMemcpyAsync(&k0, u_keys.as<uint64_t>(), 8, hipMemcpyDeviceToHost,
                         ctx->stream);
    (void)hipMemcpyAsync(&k1, u_keys.as<uint64_t>() + n_pairs - 1, 8,
                         hipMemcpyDeviceToHost, ctx->stream);
    (void)hipStreamSynchronize(ctx->stream);
    MGX_LTRACE("coarsen ukeys[0]=(%d,%d) ukeys[last]=(%d,%d)", (int)(k0 >> 32),
               (int)(uint32_t)k0, (int)(k1 >> 32), (int)(uint32_t)k1);
  }
  // CSR counts + self presence.
  DevBuf counts, self_present;
  MGX_HIP_TRY(counts.alloc(ctx, n_clusters * 4));
  MGX_HIP_TRY(self_present.alloc(ctx, n_clusters * 4));
  MGX_HIP_TRY(hipMemsetAsync(counts.p, 0, n_clusters * 4, ctx->stream));
  MGX_HIP_TRY(hipMemsetAsync(self_present.p, 0, n_clusters * 4, ctx->stream));
  if (n_pairs > 0)
    hipLaunchKernelGGL(k_pair_counts, dim3((uint32_t)grid_for(n_pairs)), dim3(kBlock), 0,
                       ctx->stream, n_pairs, u_keys.as<uint64_t>(), n_clusters,
                       counts.as<uint32_t>(), self_present.as<uint32_t>(),
                       (unsigned long long *)d_bad.p);
  {
    unsigned long long oob = 0;
    MGX_HIP_TRY(hipMemcpyAsync(&oob, d_bad.p, 8, hipMemcpyDeviceToHost, ctx->stream));
    MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
    if (oob) {
      fprintf(stderr,
              "[louvain] COARSEN GUARD: %llu out-of-range pair ids dropped "
              "(ncl=%lld) — please report\n",
              oob, (long long)n_clusters);
      fflush(stderr);
      const char *strict = getenv("MGX_LOUVAIN_STRICT");
      if (strict && atoi(strict)) {
        mgx_set_error("louvain coarsen guard fired (%llu bad pair ids)", oob);
        return MGX_ERR_HIP;
      }
    }
  }
  hipLaunchKernelGGL(k_add_missing_self, dim3((uint32_t)grid_for(n_clusters)), dim3(kBlock),
                     0, ctx->stream, n_clusters, self_present.as<uint32_t>(),
                     counts.as<uint32_t>());
  if (louvain_trace()) {
    uint32_t c0[4] = {0, 0, 0, 0};
    (void)hipMemcpyAsync(c0, counts.p, 16, hipMemcpyDeviceToHost, ctx->stream);
    hipError_t se = hipStreamSynchronize(ctx->stream);
    MGX_LTRACE("coarsen counts[0..3]=%u,%u,%u,%u sync=%s", c0[0], c0[1], c0[2], c0[3],
               hipGetErrorString(se));
  }

  out->nv = n_clusters;
  ctx->ensure_margin((size_t)(n_clusters + 1) * 4);
  MGX_HIP_TRY(mgx_hip_malloc(&out->row_ptr, (n_clusters + 1) * 4));
  if (louvain_trace()) {
    // Pre-scan probe: counts must still hold the pair histogram, and the
    // freshly hipMalloc'd row_ptr must not alias the pool-allocated counts
    // (the r02 corruption signature is an in-place exclusive scan -> zeros).
    uint32_t c1[4] = {0, 0, 0, 0};
    (void)hipMemcpyAsync(c1, counts.p, 16, hipMemcpyDeviceToHost, ctx->stream);
    (void)hipStreamSynchronize(ctx->stream);
    MGX_LTRACE("coarsen prescan counts[0..3]=%u,%u,%u,%u", c1[0], c1[1], c1[2], c1[3]);
    const char *rp0 = (const char *)out->row_ptr;
    const char *rp1 = rp0 + (size_t)(n_clusters + 1) * 4;
    const char *cn0 = (const char *)counts.p, *cn1 = cn0 + (size_t)n_clusters * 4;
    const char *ws0 = (const char *)ctx->workspace;
    const char *ws1 = ws0 + ctx->workspace_bytes;
    MGX_LTRACE("coarsen prescan row_ptr=[%p,%p) counts=[%p,%p) ws=[%p,%p)", rp0, rp1, cn0,
               cn1, ws0, ws1);
    if (rp0 < cn1 && cn0 < rp1) MGX_LTRACE("coarsen ALIAS row_ptr overlaps counts");
    if (rp0 < ws1 && ws0 < rp1) MGX_LTRACE("coarsen ALIAS row_ptr overlaps workspace");
    if (cn0 < ws1 && ws0 < cn1) MGX_LTRACE("coarsen ALIAS counts overlaps workspace");
  }
  // exclusive scan counts -> row_ptr
  err = rocprim::exclusive_scan(nullptr, tmp_bytes, counts.as<uint32_t>(), out->row_ptr,
                                0u, n_clusters, rocprim::plus<uint32_t>(), ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  MGX_TRY(ctx->reserve(tmp_bytes, &tmp));
  err = rocprim::exclusive_scan(tmp, tmp_bytes, counts.as<uint32_t>(), out->row_ptr, 0u,
                                n_clusters, rocprim::plus<uint32_t>(), ctx->stream);
  if (err != hipSuccess) return MGX_ERR_HIP;
  uint32_t last_off = 0, last_cnt = 0;
  MGX_HIP_TRY(hipMemcpyAsync(&last_off, out->row_ptr + n_clusters - 1, 4,
                             hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipMemcpyAsync(&last_cnt, counts.as<uint32_t>() + n_clusters - 1, 4,
                             hipMemcpyDeviceToHost, ctx->stream));
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  if (louvain_trace()) {
    uint32_t rp[4] = {0, 0, 0, 0}, cs[4] = {0, 0, 0, 0};
    (void)hipMemcpyAsync(rp, out->row_ptr, 16, hipMemcpyDeviceToHost, ctx->stream);
    (void)hipMemcpyAsync(cs, counts.p, 16, hipMemcpyDeviceToHost, ctx->stream);
    (void)hipStreamSynchronize(ctx->stream);
    MGX_LTRACE("coarsen scan rp[0..3]=%u,%u,%u,%u cs[0..3]=%u,%u,%u,%u last_off=%u "
               "last_cnt=%u",
               rp[0], rp[1], rp[2], rp[3], cs[0], cs[1], cs[2], cs[3], last_off,
               last_cnt);
  }
  const uint32_t total = last_off + last_cnt;
  MGX_HIP_TRY(hipMemcpyAsync(out->row_ptr + n_clusters, &total, 4, hipMemcpyHostToDevice,
                             ctx->stream));
  out->ne2 = total;
  MGX_LTRACE("coarsen n_pairs=%lld ncl=%lld total=%u", (long long)n_pairs,
             (long long)n_clusters, total);
  // Consistency gate: total = 2*offdiag + ncl >= max(n_pairs - dropped, ncl).
  // A smaller total means the counts/row_ptr memory was trampled between the
  // passes (the r02 storm signature: plain-hipMalloc row_ptr aliasing the
  // pool-allocated counts block turned the exclusive scan in-place -> all
  // zeros); scattering against it would write OOB, so fail cleanly instead.
  if ((int64_t)total < n_clusters) {
    mgx_set_error(
        "louvain coarsen inconsistent (total=%u < ncl=%lld, n_pairs=%lld) — "
        "device memory corruption detected",
        total, (long long)n_clusters, (long long)n_pairs);
    return MGX_ERR_HIP;
  }
  // col/w come from the entry-allocated ping-pong arenas when they fit
  // (they always do: total <= ne2_0 + nv_0); a direct malloc is only the
  // fallback for graphs whose first level was never arena-sized.
  const int flip = ctx->louv_flip ^= 1;
  const size_t need_c = (size_t)(total > 0 ? total : 1) * 4;
  const size_t need_w = (size_t)(total > 0 ? total : 1) * 8;
  if (ctx->louv_col[flip] && need_c <= ctx->louv_col_bytes[flip] &&
      ctx->louv_w[flip] && need_w <= ctx->louv_w_bytes[flip]) {
    out->col = (int32_t *)ctx->louv_col[flip];
    out->w = (double *)ctx->louv_w[flip];
    out->arena = true;
  } else {
    ctx->ensure_margin(need_c + need_w);
    MGX_HIP_TRY(mgx_hip_malloc(&out->col, need_c));
    MGX_HIP_TRY(mgx_hip_malloc(&out->w, need_w));
  }
  {
    // ALWAYS-ON overlap gate. The box's ROCm 7.0.x runtime mis-places
    // multi-GiB hipMallocs once the VA space is fragmented: at RMAT-25+
    // the total-sized col/w come back overlapping five live buffers (the
    // fit check appears to use only the low 32 bits of the size; an
    // isolated fresh-process repro allocates the same sizes correctly).
    // Scattering through such a mapping faults the GPU, so detect the
    // overlap host-side and fail cleanly instead.
    struct Rg {
      const void *p;
      size_t bytes;
    } rgs[] = {
        {counts.p, (size_t)n_clusters * 4},
        {self_present.p, (size_t)n_clusters * 4},
        {row_of_entry.p, (size_t)ne2 * 4},
        {keys.p, (size_t)ne2 * 8},
        {vals.p, (size_t)ne2 * 8},
        {keys_sorted.p, (size_t)ne2 * 8},
        {vals_sorted.p, (size_t)ne2 * 8},
        {u_keys.p, (size_t)ne2 * 8},
        {u_vals.p, (size_t)ne2 * 8},
        {ctx->workspace, ctx->workspace_bytes},
        {in.row_ptr, (size_t)(in.nv + 1) * 4},
        {in.col, (size_t)ne2 * 4},
        {in.w, (size_t)ne2 * 8},
        {out->row_ptr, (size_t)(n_clusters + 1) * 4},
        {out->col, (size_t)(total > 0 ? total : 1) * 4},
        {out->w, (size_t)(total > 0 ? total : 1) * 8},
    };
    const size_t nr = sizeof(rgs) / sizeof(rgs[0]);
    for (size_t a = 0; a < nr; ++a)
      for (size_t b = a + 1; b < nr; ++b) {
        const char *a0 = (const char *)rgs[a].p, *a1 = a0 + rgs[a].bytes;
        const char *b0 = (const char *)rgs[b].p, *b1 = b0 + rgs[b].bytes;
        if (a0 && b0 && a0 < b1 && b0 < a1) {
          mgx_set_error(
              "louvain coarsen: the runtime returned overlapping device "
              "allocations (ranges %zu and %zu; ncl=%lld total=%u) — known "
              "ROCm 7.0.x VA-placement fault at this scale, see DESIGN.md",
              a, b, (long long)n_clusters, total);
          return MGX_ERR_HIP;
        }
      }
  }
  if (louvain_trace()) {
    // Allocator-overlap probe: the async-pool DevBufs vs the plain hipMalloc
    // level buffers must be disjoint VA ranges.
    struct Rng {
      const char *name;
      const void *p;
      size_t bytes;
    } rngs[] = {
        {"counts", counts.p, (size_t)n_clusters * 4},
        {"self_present", self_present.p, (size_t)n_clusters * 4},
        {"row_of_entry", row_of_entry.p, (size_t)ne2 * 4},
        {"keys", keys.p, (size_t)ne2 * 8},
        {"vals", vals.p, (size_t)ne2 * 8},
        {"keys_sorted", keys_sorted.p, (size_t)ne2 * 8},
        {"vals_sorted", vals_sorted.p, (size_t)ne2 * 8},
        {"u_keys", u_keys.p, (size_t)ne2 * 8},
        {"u_vals", u_vals.p, (size_t)ne2 * 8},
        {"workspace", ctx->workspace, ctx->workspace_bytes},
        {"in.row_ptr", in.row_ptr, (size_t)(in.nv + 1) * 4},
        {"in.col", in.col, (size_t)ne2 * 4},
        {"in.w", in.w, (size_t)ne2 * 8},
        {"row_ptr", out->row_ptr, (size_t)(n_clusters + 1) * 4},
        {"col", out->col, (size_t)(total > 0 ? total : 1) * 4},
        {"w", out->w, (size_t)(total > 0 ? total : 1) * 8},
    };
    for (auto &r : rngs)
      MGX_LTRACE("coarsen buf %-12s [%p, %p)", r.name, r.p,
                 (const void *)((const char *)r.p + r.bytes));
    for (size_t a = 0; a < sizeof(rngs) / sizeof(rngs[0]); ++a)
      for (size_t b = a + 1; b < sizeof(rngs) / sizeof(rngs[0]); ++b) {
        const char *a0 = (const char *)rngs[a].p, *a1 = a0 + rngs[a].bytes;
        const char *b0 = (const char *)rngs[b].p, *b1 = b0 + rngs[b].bytes;
        if (a0 < b1 && b0 < a1)
          MGX_LTRACE("coarsen ALIAS %s overlaps %s", rngs[a].name, rngs[b].name);
      }
  }
  // cursor = row_ptr copy (reuse counts buffer)
  MGX_HIP_TRY(hipMemcpyAsync(counts.p, out->row_ptr, n_clusters * 4,
                             hipMemcpyDeviceToDevice, ctx->stream));
  MGX_LTRACE("coarsen pair_scatter");
  if (n_pairs > 0)
    hipLaunchKernelGGL(k_pair_scatter, dim3((uint32_t)grid_for(n_pairs)), dim3(kBlock), 0,
                       ctx->stream, n_pairs, u_keys.as<uint64_t>(), u_vals.as<double>(),
                       n_clusters, total, counts.as<uint32_t>(), out->col, out->w);
  MGX_LTRACE("coarsen self_scatter");
  hipLaunchKernelGGL(k_self_scatter, dim3((uint32_t)grid_for(n_clusters)), dim3(kBlock), 0,
                     ctx->stream, n_clusters, self_present.as<uint32_t>(), total,
                     counts.as<uint32_t>(), out->col, out->w);
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  MGX_HIP_TRY(hipGetLastError());
  MGX_LTRACE("coarsen done");
  return MGX_OK;
}

void free_level(Level *L, bool own) {
  if (!own) return;
  if (L->row_ptr) (void)hipFree(L->row_ptr);
  if (!L->arena) {
    if (L->col) (void)hipFree(L->col);
    if (L->w) (void)hipFree(L->w);
  }
  L->row_ptr = nullptr;
  L->col = nullptr;
  L->w = nullptr;
}

__global__ void k_compose(int64_t nv, int32_t *c_orig, const int32_t *C) {
  // runMultiPhaseBasic.cpp:101-106: C_orig[i] = C[C_orig[i]] when >= 0.
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    if (c_orig[i] >= 0) c_orig[i] = C[c_orig[i]];
  }
}

__global__ void k_widen_i32_to_i64(int64_t n, const int32_t *in, int64_t *out) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x)
    out[i] = (int64_t)in[i];
}

}  // namespace

mgx_status mgx_louvain_impl(mgx_context *ctx, mgx_graph *g, double threshold,
                            int64_t *out_community, int64_t *n_communities) {
  if (!(g->flags & MGX_BUILD_SYM_CSR)) {
    mgx_set_error("louvain needs a graph built with MGX_BUILD_SYM_CSR");
    return MGX_ERR_INVALID_ARGUMENT;
  }
  const int64_t nv0 = g->n_vertices;
  if (n_communities) *n_communities = 0;
  if (out_community) {
    for (int64_t i = 0; i < nv0; ++i) out_community[i] = -1;
  }
  if (nv0 == 0 || g->n_edges == 0) return MGX_OK;

  // Level 0: widen weights to fp64 (unweighted graphs: all 1.0).
  Level L;
  L.nv = nv0;
  L.row_ptr = g->sym_row_ptr;
  L.col = g->sym_col;
  L.ne2 = 2 * g->n_edges;
  // Size the coarsen col/w arenas now, before any per-level VA churn: a
  // coarse level has at most ne2 + nv entries (every off-diagonal pair
  // twice + one self loop per cluster).
  {
    const size_t bound = (size_t)(L.ne2 + nv0 + 1);
    for (int i = 0; i < 2; ++i) {
      if (ctx->louv_col_bytes[i] < bound * 4) {
        if (ctx->louv_col[i]) (void)hipFree(ctx->louv_col[i]);
        ctx->louv_col[i] = nullptr;
        ctx->louv_col_bytes[i] = 0;
        MGX_HIP_TRY(mgx_hip_malloc(&ctx->louv_col[i], bound * 4));
        ctx->louv_col_bytes[i] = bound * 4;
      }
      if (ctx->louv_w_bytes[i] < bound * 8) {
        if (ctx->louv_w[i]) (void)hipFree(ctx->louv_w[i]);
        ctx->louv_w[i] = nullptr;
        ctx->louv_w_bytes[i] = 0;
        MGX_HIP_TRY(mgx_hip_malloc(&ctx->louv_w[i], bound * 8));
        ctx->louv_w_bytes[i] = bound * 8;
      }
    }
  }
  DevBuf w0;
  MGX_HIP_TRY(w0.alloc(ctx, L.ne2 * 8));
  if (g->sym_w) {
    hipLaunchKernelGGL(k_f32_to_f64, dim3((uint32_t)grid_for(L.ne2)), dim3(kBlock), 0,
                       ctx->stream, L.ne2, g->sym_w, (double *)w0.p);
  } else {
    struct K {
      static __global__ void fill1(int64_t n, double *p) {
        for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
             i += (int64_t)gridDim.x * blockDim.x)
          p[i] = 1.0;
      }
    };
    hipLaunchKernelGGL(K::fill1, dim3((uint32_t)grid_for(L.ne2)), dim3(kBlock), 0,
                       ctx->stream, L.ne2, (double *)w0.p);
  }
  L.w = (double *)w0.p;
  bool own_level = false;

  DevBuf c_orig_d, c_level;
  MGX_HIP_TRY(c_orig_d.alloc(ctx, nv0 * 4));
  hipLaunchKernelGGL(k_fill_i32, dim3((uint32_t)grid_for(nv0)), dim3(kBlock), 0,
                     ctx->stream, nv0, -1, c_orig_d.as<int32_t>());

  double prev_mod = -1.0, curr_mod = -1.0;
  int64_t phase = 1, tot_itr = 0;
  mgx_status status = MGX_OK;
  while (true) {
    DevBuf C;
    if (C.alloc(ctx, L.nv * 4) != hipSuccess) {
      status = MGX_ERR_OUT_OF_MEMORY;
      break;
    }
    prev_mod = curr_mod;
    int64_t iters = 0;
    MGX_LDBG(ctx, "phase=%lld level nv=%lld ne2=%lld", (long long)phase, (long long)L.nv,
             (long long)L.ne2);
    status = louvain_level(ctx, L, prev_mod, threshold, C.as<int32_t>(), &curr_mod, &iters);
    if (status != MGX_OK) break;
    tot_itr += iters;
    int64_t n_clusters = 0;
    status = renumber(ctx, C.as<int32_t>(), L.nv, &n_clusters);
    if (status != MGX_OK) break;
    if (phase == 1) {
      MGX_HIP_TRY(hipMemcpyAsync(c_orig_d.p, C.p, nv0 * 4, hipMemcpyDeviceToDevice,
                                 ctx->stream));
    } else {
      hipLaunchKernelGGL(k_compose, dim3((uint32_t)grid_for(nv0)), dim3(kBlock), 0,
                         ctx->stream, nv0, c_orig_d.as<int32_t>(), C.as<int32_t>());
    }
    if (phase > 200 || tot_itr > 100000) break;
    if ((curr_mod - prev_mod) > threshold) {
      Level next;
      MGX_LDBG(ctx, "coarsen from nv=%lld to %lld", (long long)L.nv,
               (long long)n_clusters);
      status = coarsen(ctx, L, C.as<int32_t>(), n_clusters, &next);
      free_level(&L, own_level);
      if (status != MGX_OK) break;
      L = next;
      own_level = true;
      ++phase;
    } else {
      break;
    }
  }
  free_level(&L, own_level);
  if (status != MGX_OK) return status;

  // Download.
  DevBuf wide;
  MGX_HIP_TRY(wide.alloc(ctx, nv0 * 8));
  hipLaunchKernelGGL(k_widen_i32_to_i64, dim3((uint32_t)grid_for(nv0)), dim3(kBlock), 0,
                     ctx->stream, nv0, c_orig_d.as<int32_t>(), wide.as<int64_t>());
  if (out_community) {
    MGX_HIP_TRY(hipMemcpyAsync(out_community, wide.p, nv0 * 8, hipMemcpyDeviceToHost,
                               ctx->stream));
  }
  MGX_HIP_TRY(hipStreamSynchronize(ctx->stream));
  if (n_communities && out_community) {
    int64_t mx = -1;
    for (int64_t i = 0; i < nv0; ++i)
      if (out_community[i] > mx) mx = out_community[i];
    *n_communities = mx + 1;
  }
  MGX_HIP_TRY(hipGetLastError());
  return MGX_OK;
}
