// Deterministic synthetic graph generation shared by every implementation
// (oracle C++, HIP kernels, and the numpy port in memgraph_amd/rmat.py).
//
// All three implementations must agree BIT-FOR-BIT so the CPU oracle and the
// GPU path can generate the same graph independently (no network, no
// datasets — BASELINE.md workloads are synthetic with fixed seeds).
//
// The edge stream is counter-based: edge i is a pure function of
// (seed, i), so any contiguous sub-range can be generated anywhere
// (host, device, any rank) without materializing the rest.
//
// RMAT parameters follow the reference's own generator defaults
// (Graph500 A/B/C/D = .57/.19/.19/.05, clip_and_flip off, multi-edges and
// self-loops kept — /root/reference/src/mage/cpp/cugraph_module/algorithms/
// graph_generator.cu:143-147 and pagerank.hpp:27 allow multi-edges).

#ifndef MGX_GRAPHGEN_H
#define MGX_GRAPHGEN_H

#include <stdint.h>

#ifdef __HIPCC__
#define MGX_HD __host__ __device__
#else
#define MGX_HD
#endif

// splitmix64 finalizer (public-domain mixing constants).
MGX_HD static inline uint64_t mgx_mix64(uint64_t x) {
  x ^= x >> 30;
  x *= 0xBF58476D1CE4E5B9ULL;
  x ^= x >> 27;
  x *= 0x94D049BB133111EBULL;
  x ^= x >> 31;
  return x;
}

MGX_HD static inline uint64_t mgx_seed_mix(uint64_t seed) {
  return mgx_mix64(seed ^ 0x5851F42D4C957F2DULL);
}

// Counter-based stream: value idx of stream `mixed_seed`.
MGX_HD static inline uint64_t mgx_hash64(uint64_t mixed_seed, uint64_t idx) {
  return mgx_mix64(mixed_seed + idx * 0x9E3779B97F4A7C15ULL);
}

// RMAT quadrant thresholds as u64 (integer compares everywhere — no float
// divergence between host/device/numpy).
typedef struct {
  uint64_t t_a, t_ab, t_abc;
} mgx_rmat_thresholds;

MGX_HD static inline mgx_rmat_thresholds mgx_rmat_make_thresholds(double a, double b, double c) {
  mgx_rmat_thresholds t;
  const double two64 = 18446744073709551616.0;
  t.t_a = (uint64_t)(a * two64);
  t.t_ab = (uint64_t)((a + b) * two64);
  t.t_abc = (uint64_t)((a + b + c) * two64);
  return t;
}

// Edge i of an RMAT(scale) graph: V = 2^scale. Bit for level l is placed at
// bit position l (LSB-first).
MGX_HD static inline void mgx_rmat_edge(uint64_t mixed_seed, uint64_t i, int scale,
                                        mgx_rmat_thresholds t, uint64_t *src, uint64_t *dst) {
  uint64_t s = 0, d = 0;
  for (int l = 0; l < scale; ++l) {
    uint64_t h = mgx_hash64(mixed_seed, i * (uint64_t)scale + (uint64_t)l);
    uint64_t rb, cb;
    if (h < t.t_a) {
      rb = 0; cb = 0;
    } else if (h < t.t_ab) {
      rb = 0; cb = 1;
    } else if (h < t.t_abc) {
      rb = 1; cb = 0;
    } else {
      rb = 1; cb = 1;
    }
    s |= rb << l;
    d |= cb << l;
  }
  *src = s;
  *dst = d;
}

// Edge i of a uniform random directed graph on V vertices.
MGX_HD static inline void mgx_uniform_edge(uint64_t mixed_seed, uint64_t i, uint64_t n_vertices,
                                           uint64_t *src, uint64_t *dst) {
  *src = mgx_hash64(mixed_seed, 2 * i) % n_vertices;
  *dst = mgx_hash64(mixed_seed, 2 * i + 1) % n_vertices;
}

// Weight of edge i in [0, 1): 53-bit mantissa, identical to
// (hash >> 11) * 2^-53 in numpy.
MGX_HD static inline double mgx_edge_weight(uint64_t mixed_wseed, uint64_t i) {
  return (double)(mgx_hash64(mixed_wseed, i) >> 11) * (1.0 / 9007199254740992.0);
}

#endif  // MGX_GRAPHGEN_H
