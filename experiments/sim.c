// Cache simulator for the PageRank fused pull-SpMV sweep on MI355X
// (experiment tooling, CPU-only; not part of the product).
//
// Models the sweep kernel's gather traffic through the chip's cache
// hierarchy to rank vertex-ordering candidates offline before spending GPU
// minutes: 8 per-XCD L2s (4 MiB, 16-way, 64 B lines, not shared) in front
// of one die-level 256 MiB 16-way L3, blocks round-robin to XCDs (b % 8),
// degree-binned launch geometry identical to memgraph_amd/csrc/pagerank.hip
// (bins {<8,<64,<1024,>=1024} at {4,16,64,256} lanes/row, grid caps
// {2048,2048,2048,8192}, grid-stride), block time-interleaved by a fixed
// per-step edge quantum.
//
// stdin/argv: sim <row_ptr.bin u32[V+1]> <col.bin i32[E]> <V> [--no-colstream]
// Outputs fabric (L2-miss) and HBM (L3-miss) gather bytes per sweep, plus
// stream traffic accounting.
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>

typedef struct {
  uint32_t *tags;  // [sets][ways]
  uint8_t *age;    // LRU ages
  int sets, ways;
  uint64_t hits, misses;
} Cache;

static void cache_init(Cache *c, long bytes, int ways) {
  c->ways = ways;
  c->sets = (int)(bytes / 64 / ways);
  c->tags = malloc((size_t)c->sets * ways * 4);
  c->age = malloc((size_t)c->sets * ways);
  memset(c->tags, 0xFF, (size_t)c->sets * ways * 4);
  memset(c->age, 0, (size_t)c->sets * ways);
  c->hits = c->misses = 0;
}

// returns 1 on hit
static inline int cache_access(Cache *c, uint64_t line) {
  const int set = (int)(line & (uint64_t)(c->sets - 1));
  uint32_t tag = (uint32_t)(line / c->sets);
  uint32_t *t = c->tags + (size_t)set * c->ways;
  uint8_t *a = c->age + (size_t)set * c->ways;
  int victim = 0;
  uint8_t worst = 0;
  for (int w = 0; w < c->ways; ++w) {
    if (t[w] == tag) {
      c->hits++;
      // LRU: age of hit way -> 0, younger ways age up
      uint8_t old = a[w];
      for (int x = 0; x < c->ways; ++x)
        if (a[x] < old) a[x]++;
      a[w] = 0;
      return 1;
    }
    if (a[w] >= worst) { worst = a[w]; victim = w; }
  }
  c->misses++;
  t[victim] = tag;
  for (int x = 0; x < c->ways; ++x)
    if (x != victim) { if (a[x] < 255) a[x]++; }
  a[victim] = 0;
  return 0;
}

typedef struct {
  // a block's work: list of rows (by index range into a bin row list), and
  // a cursor (current row position + intra-row column offset)
  int bin;
  long blk_in_sec;
  long next_base;   // row-list base of next group
  long cur_row_i;   // index into bin rows for in-progress big row
  uint32_t cur_col; // next col offset for in-progress big row
  int done;
} Block;

int main(int argc, char **argv) {
  if (argc < 4) { fprintf(stderr, "usage: sim row_ptr.bin col.bin V [--no-colstream] [binprefix]\n"); return 2; }
  int colstream = 1;
  const char *binprefix = NULL;
  for (int a = 4; a < argc; ++a) {
    if (!strcmp(argv[a], "--no-colstream")) colstream = 0;
    else binprefix = argv[a];
  }
  long V = atol(argv[3]);
  FILE *f = fopen(argv[1], "rb");
  uint32_t *row_ptr = malloc((V + 1) * 4);
  if (fread(row_ptr, 4, V + 1, f) != (size_t)(V + 1)) return 3;
  fclose(f);
  long E = row_ptr[V];
  int32_t *col = malloc(E * 4);
  f = fopen(argv[2], "rb");
  if (fread(col, 4, E, f) != (size_t)E) return 3;
  fclose(f);

  // degree bins (mirrors k_bin_keys + launch geometry in the product)
  long cnt[4] = {0, 0, 0, 0};
  for (long r = 0; r < V; ++r) {
    uint32_t d = row_ptr[r + 1] - row_ptr[r];
    int k = d < 8 ? 0 : d < 64 ? 1 : d < 1024 ? 2 : 3;
    cnt[k]++;  // include_zero=true in the product build for bins_in
  }
  int32_t *rows[4];
  for (int b = 0; b < 4; ++b) rows[b] = malloc((cnt[b] ? cnt[b] : 1) * 4);
  if (binprefix) {
    // custom bin row lists (position-aware XCD clustering experiments)
    for (int b = 0; b < 4; ++b) {
      char p[512];
      snprintf(p, sizeof p, "%s.rows%d.bin", binprefix, b);
      FILE *bf = fopen(p, "rb");
      if (!bf) { fprintf(stderr, "missing %s\n", p); return 4; }
      if (cnt[b] > 0 && fread(rows[b], 4, cnt[b], bf) != (size_t)cnt[b]) return 5;
      fclose(bf);
    }
  } else {
  long fill[4] = {0, 0, 0, 0};
  for (long r = 0; r < V; ++r) {
    uint32_t d = row_ptr[r + 1] - row_ptr[r];
    int k = d < 8 ? 0 : d < 64 ? 1 : d < 1024 ? 2 : 3;
    rows[k][fill[k]++] = (int32_t)r;
  }
  }
  const long rpb[4] = {64, 16, 4, 1};
  const long cap[4] = {2048, 2048, 2048, 8192};
  long grid[4];
  for (int b = 0; b < 4; ++b) {
    long need = (cnt[b] + rpb[b] - 1) / rpb[b];
    grid[b] = cnt[b] ? (need < cap[b] ? need : cap[b]) : 0;
  }
  long nblocks = grid[0] + grid[1] + grid[2] + grid[3];
  Block *blk = malloc(nblocks * sizeof(Block));
  long bi = 0;
  for (int b = 0; b < 4; ++b)
    for (long g = 0; g < grid[b]; ++g, ++bi) {
      blk[bi].bin = b;
      blk[bi].blk_in_sec = g;
      blk[bi].next_base = g * rpb[b];
      blk[bi].cur_row_i = -1;
      blk[bi].cur_col = 0;
      blk[bi].done = 0;
    }

  Cache l2[8], l3;
  for (int x = 0; x < 8; ++x) cache_init(&l2[x], 4l << 20, 16);
  cache_init(&l3, 256l << 20, 16);
  uint64_t col_fabric = 0;  // col-stream lines through L2 (nt = L2-served)

  const long QUANTUM = 2048;  // edges per block per scheduling step
  long active = nblocks;
  while (active > 0) {
    active = 0;
    for (long i = 0; i < nblocks; ++i) {
      Block *B = &blk[i];
      if (B->done) continue;
      int xcd = (int)(i & 7);
      long budget = QUANTUM;
      while (budget > 0) {
        if (B->cur_row_i < 0) {
          // start next row group
          if (B->next_base >= cnt[B->bin]) { B->done = 1; break; }
          B->cur_row_i = B->next_base;
          B->cur_col = 0;
        }
        long group_end = B->cur_row_i == -1 ? 0 : (B->next_base + rpb[B->bin]);
        // process rows of the current group until budget exhausted
        while (B->cur_row_i < group_end && B->cur_row_i < cnt[B->bin] && budget > 0) {
          int32_t r = rows[B->bin][B->cur_row_i];
          uint32_t s = row_ptr[r] + B->cur_col, e = row_ptr[r + 1];
          uint32_t take = e - s;
          if ((long)take > budget) take = (uint32_t)budget;
          for (uint32_t j = s; j < s + take; ++j) {
            uint64_t line = (uint64_t)(col[j] >> 4);
            if (!cache_access(&l2[xcd], line)) cache_access(&l3, line);
            if (colstream) {
              // col stream: 16 cols per 64-B line, sequential; model only
              // the line-granular fetch into the same L2 (pollution)
              if ((j & 15u) == 0) {
                uint64_t cline = 0x80000000ull + (j >> 4);  // distinct space
                if (!cache_access(&l2[xcd], cline)) { cache_access(&l3, cline); }
                col_fabric++;
              }
            }
          }
          budget -= take;
          B->cur_col += take;
          if (B->cur_col >= row_ptr[r + 1] - row_ptr[r]) {
            B->cur_row_i++;
            B->cur_col = 0;
          }
        }
        if (B->cur_row_i >= group_end || B->cur_row_i >= cnt[B->bin]) {
          // group finished: grid-stride to next
          B->next_base += grid[B->bin] * rpb[B->bin];
          B->cur_row_i = -1;
          if (B->next_base >= cnt[B->bin]) { B->done = 1; break; }
        }
      }
      if (!B->done) active++;
    }
  }

  uint64_t l2h = 0, l2m = 0;
  for (int x = 0; x < 8; ++x) { l2h += l2[x].hits; l2m += l2[x].misses; }
  // separate gather vs col-stream accounting is approximate: both share L2
  double fabric_gather_gb = 0, hbm_gb = l3.misses * 64.0 / 1e9;
  fabric_gather_gb = (l2m)*64.0 / 1e9;
  double vertex_gb = V * 20.0 / 1e9;
  printf("V=%ld E=%ld blocks=%ld bins=[%ld,%ld,%ld,%ld]\n", V, E, nblocks,
         cnt[0], cnt[1], cnt[2], cnt[3]);
  printf("L2: hits=%llu misses=%llu hit_rate=%.3f\n",
         (unsigned long long)l2h, (unsigned long long)l2m,
         (double)l2h / (double)(l2h + l2m));
  printf("L3: hits=%llu misses=%llu hit_rate=%.3f\n",
         (unsigned long long)l3.hits, (unsigned long long)l3.misses,
         (double)l3.hits / (double)(l3.hits + l3.misses + 1));
  printf("fabric (L2-miss) bytes/sweep: %.2f GB (+%.2f GB vertex streams)\n",
         fabric_gather_gb, vertex_gb);
  printf("HBM (L3-miss) bytes/sweep:    %.2f GB\n", hbm_gb);
  return 0;
}
