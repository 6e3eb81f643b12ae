// Experiment tooling (CPU-only): generate RMAT edges with the repo's
// deterministic generator, apply a candidate vertex ordering, build the
// permuted sorted in-CSR exactly as graph_build.hip does (sort by
// (perm[dst]<<32)|perm[src]), and write row_ptr.bin/col.bin for sim.c.
//
//   ordergen <scale> <mode> <outprefix>
// modes: deg_desc | hot4_mindst | hot4_natural
#include <parallel/algorithm>

#include <algorithm>
#include <cstdint>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <vector>

#include "../include/mgx_graphgen.h"

int main(int argc, char **argv) {
  if (argc < 4) {
    fprintf(stderr, "usage: ordergen <scale> <mode> <outprefix>\n");
    return 2;
  }
  const int scale = atoi(argv[1]);
  const char *mode = argv[2];
  const char *out = argv[3];
  const int64_t V = 1ll << scale;
  const int64_t E = 16 * V;
  const uint64_t ms = mgx_seed_mix(1);
  const mgx_rmat_thresholds t = mgx_rmat_make_thresholds(0.57, 0.19, 0.19);

  std::vector<int32_t> src(E), dst(E);
#pragma omp parallel for
  for (int64_t i = 0; i < E; ++i) {
    uint64_t s, d;
    mgx_rmat_edge(ms, (uint64_t)i, scale, t, &s, &d);
    src[i] = (int32_t)s;
    dst[i] = (int32_t)d;
  }
  std::vector<uint32_t> outdeg(V, 0);
  for (int64_t i = 0; i < E; ++i) outdeg[src[i]]++;

  // deg-desc stable order (ties by id) — matches graph_build.hip's
  // radix_sort_pairs_desc on (outdeg, iota).
  std::vector<int32_t> order(V);
  for (int64_t v = 0; v < V; ++v) order[v] = (int32_t)v;
  __gnu_parallel::stable_sort(order.begin(), order.end(),
                              [&](int32_t a, int32_t b) { return outdeg[a] > outdeg[b]; });

  std::vector<int32_t> perm(V);
  if (!strcmp(mode, "deg_desc")) {
    for (int64_t k = 0; k < V; ++k) perm[order[k]] = (int32_t)k;
  } else {
    // hot prefix = top 1M by degree (4 MB of f32 contrib)
    const int64_t K = 1 << 20;
    std::vector<int32_t> permdd(V);
    for (int64_t k = 0; k < V; ++k) permdd[order[k]] = (int32_t)k;
    std::vector<int64_t> key(V);
    if (!strcmp(mode, "hot4_mindst")) {
      // tail key: smallest deg-desc-permuted destination that gathers the
      // source, so a destination row's tail gathers become contiguous runs
      std::vector<int32_t> mindst(V, INT32_MAX);
      for (int64_t i = 0; i < E; ++i) {
        const int32_t pd = permdd[dst[i]];
        if (pd < mindst[src[i]]) mindst[src[i]] = pd;
      }
      for (int64_t v = 0; v < V; ++v) key[v] = mindst[v];
    } else {  // hot4_natural: tail in original-id order
      for (int64_t v = 0; v < V; ++v) key[v] = v;
    }
    std::vector<int32_t> tail;
    tail.reserve(V - K);
    std::vector<uint8_t> is_hot(V, 0);
    for (int64_t k = 0; k < K; ++k) is_hot[order[k]] = 1;
    for (int64_t v = 0; v < V; ++v)
      if (!is_hot[v]) tail.push_back((int32_t)v);
    __gnu_parallel::stable_sort(tail.begin(), tail.end(), [&](int32_t a, int32_t b) {
      if (key[a] != key[b]) return key[a] < key[b];
      return outdeg[a] > outdeg[b];
    });
    for (int64_t k = 0; k < K; ++k) perm[order[k]] = (int32_t)k;
    for (size_t k = 0; k < tail.size(); ++k) perm[tail[k]] = (int32_t)(K + k);
  }

  // hot4_xcd3: like hot4 but the tail orders by (top-3 original-id bits,
  // degree desc) — the RMAT bit-prefix correlation groups sources gathered
  // by same-prefix destinations — and bin row lists are written with
  // POSITION-AWARE XCD filling: rows whose original id shares a top-3
  // prefix land on list positions whose covering block maps to the same
  // XCD (b % 8), so each XCD's 4 MB L2 caches its own source cluster.
  const bool xcd_mode = !strcmp(mode, "hot4_xcd3");
  if (xcd_mode) {
    const int64_t K = 1 << 20;
    std::vector<int32_t> permdd(V);
    for (int64_t k = 0; k < V; ++k) permdd[order[k]] = (int32_t)k;
    std::vector<uint8_t> is_hot(V, 0);
    for (int64_t k = 0; k < K; ++k) is_hot[order[k]] = 1;
    std::vector<int32_t> tail;
    tail.reserve(V - K);
    for (int64_t v = 0; v < V; ++v)
      if (!is_hot[v]) tail.push_back((int32_t)v);
    const int shift = scale - 3;
    __gnu_parallel::stable_sort(tail.begin(), tail.end(), [&](int32_t a, int32_t b) {
      const int ca = a >> shift, cb = b >> shift;
      if (ca != cb) return ca < cb;
      return outdeg[a] > outdeg[b];
    });
    for (int64_t k = 0; k < K; ++k) perm[order[k]] = (int32_t)k;
    for (size_t k = 0; k < tail.size(); ++k) perm[tail[k]] = (int32_t)(K + k);
  }

  // permuted sorted CSR
  std::vector<uint64_t> keys(E);
#pragma omp parallel for
  for (int64_t i = 0; i < E; ++i)
    keys[i] = ((uint64_t)(uint32_t)perm[dst[i]] << 32) | (uint32_t)perm[src[i]];
  src.clear();
  src.shrink_to_fit();
  dst.clear();
  dst.shrink_to_fit();
  __gnu_parallel::sort(keys.begin(), keys.end());

  std::vector<uint32_t> row_ptr(V + 1, 0);
  for (int64_t i = 0; i < E; ++i) row_ptr[(keys[i] >> 32) + 1]++;
  for (int64_t v = 0; v < V; ++v) row_ptr[v + 1] += row_ptr[v];
  std::vector<int32_t> col(E);
#pragma omp parallel for
  for (int64_t i = 0; i < E; ++i) col[i] = (int32_t)(uint32_t)keys[i];

  if (xcd_mode) {
    // orig id per permuted row
    std::vector<int32_t> orig(V);
    for (int64_t v = 0; v < V; ++v) orig[perm[v]] = (int32_t)v;
    const int shift = scale - 3;
    const long rpb[4] = {64, 16, 4, 1};
    const long capb[4] = {2048, 2048, 2048, 8192};
    for (int b = 0; b < 4; ++b) {
      std::vector<int32_t> binrows;
      for (int64_t r = 0; r < V; ++r) {
        const uint32_t d = row_ptr[r + 1] - row_ptr[r];
        const int k = d < 8 ? 0 : d < 64 ? 1 : d < 1024 ? 2 : 3;
        if (k == b) binrows.push_back((int32_t)r);
      }
      const long cnt = (long)binrows.size();
      long need = (cnt + rpb[b] - 1) / rpb[b];
      const long grid = cnt ? (need < capb[b] ? need : capb[b]) : 0;
      // cluster queues (rows in ascending permuted id within cluster)
      std::vector<std::vector<int32_t>> q(8);
      for (auto r : binrows) q[orig[r] >> shift].push_back(r);
      std::vector<size_t> head(8, 0);
      std::vector<int32_t> filled(cnt);
      const long seg = grid * rpb[b];
      for (long p = 0; p < cnt; ++p) {
        const int x = (int)(((p % (seg > 0 ? seg : 1)) / rpb[b]) % 8);
        int pick = -1;
        if (head[x] < q[x].size()) {
          pick = x;
        } else {  // spill: take from the fullest remaining cluster
          size_t best = 0;
          for (int c = 0; c < 8; ++c) {
            const size_t rem = q[c].size() - head[c];
            if (rem > best) { best = rem; pick = c; }
          }
        }
        filled[p] = q[pick][head[pick]++];
      }
      char bp[512];
      snprintf(bp, sizeof bp, "%s.rows%d.bin", out, b);
      FILE *bf = fopen(bp, "wb");
      fwrite(filled.data(), 4, cnt, bf);
      fclose(bf);
    }
  }

  char path[512];
  snprintf(path, sizeof path, "%s.row_ptr.bin", out);
  FILE *f = fopen(path, "wb");
  fwrite(row_ptr.data(), 4, V + 1, f);
  fclose(f);
  snprintf(path, sizeof path, "%s.col.bin", out);
  f = fopen(path, "wb");
  fwrite(col.data(), 4, E, f);
  fclose(f);
  printf("wrote %s (V=%lld E=%lld)\n", out, (long long)V, (long long)E);
  return 0;
}
