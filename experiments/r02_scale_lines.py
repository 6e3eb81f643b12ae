"""Round-2 scale datapoints for the SURVEY.md §8f rows (online modules,
Leiden, LabelRankT) at RMAT-22, plus optional Louvain at a given scale.

Usage: python experiments/r02_scale_lines.py [algo ...]
  algos: leiden lrt pronline konline louvain26  (default: all but louvain26)
"""
import sys, time
import numpy as np

sys.path.insert(0, "/root/repo")
from memgraph_amd.native import (BUILD_IN_CSR, BUILD_NO_PERM, BUILD_OUT_CSR,
                                 BUILD_SYM_CSR, BUILD_WEIGHTED, Native)

algos = sys.argv[1:] or ["leiden", "lrt", "pronline", "konline"]
n = Native()
ctx = n.init(0)
import os
SCALE = int(os.environ.get("MGX_LINES_SCALE", "22"))
V = 1 << SCALE
E = 16 * V


def bench(name, flags, fn):
    g = n.graph_rmat(ctx, SCALE, E, seed=1, flags=flags)
    print(f"[{name}] graph built ({n.graph_build_ms(g):.0f} ms)", flush=True)
    t0 = time.perf_counter()
    out = fn(g)
    t1 = time.perf_counter()
    print(f"[{name}] {t1 - t0:.2f} s :: {out}", flush=True)
    n.graph_destroy(ctx, g)


ids = np.arange(V, dtype=np.int64)

if "leiden" in algos:
    LEIDEN_ITERS = int(os.environ.get("MGX_LINES_LEIDEN_ITERS", str(1 << 62)))
    def run_leiden(g):
        hier, levels = n.leiden(ctx, g, V, seed=42, max_iterations=LEIDEN_ITERS)
        top = hier[np.arange(V), levels - 1]
        return f"levels max={levels.max()}, top communities={len(np.unique(top))}"
    bench(f"leiden RMAT-{SCALE}w", BUILD_SYM_CSR | BUILD_WEIGHTED, run_leiden)

if "lrt" in algos:
    # LabelRankT seeds deg+1 labels per node (the reference's init), so the
    # first propagate is Theta(sum deg^2): infeasible on power-law RMAT for
    # any implementation. Measure on a bounded-degree uniform graph.
    def run_lrt(g):
        comm = n.lrt_set(ctx, g, ids, directed=False, weighted=True)
        k = len(np.unique(comm))
        n.lrt_reset(ctx)
        return f"{k} communities"
    g = n.graph_uniform(ctx, V, E, seed=42, flags=BUILD_SYM_CSR | BUILD_WEIGHTED)
    print(f"[labelrankt UNIFORM-{SCALE}w] graph built ({n.graph_build_ms(g):.0f} ms)",
          flush=True)
    import time as _t
    _t0 = _t.perf_counter()
    out = run_lrt(g)
    print(f"[labelrankt UNIFORM-{SCALE}w] {_t.perf_counter()-_t0:.2f} s :: {out}",
          flush=True)
    n.graph_destroy(ctx, g)

if "pronline" in algos:
    def run_pronline(g):
        t0 = time.perf_counter()
        n.pronline_set(ctx, g, ids, R=10, eps=0.2, seed=1)
        t_set = time.perf_counter() - t0
        t0 = time.perf_counter()
        ranks, _consistent = n.pronline_get(ctx, ids)
        t_get = time.perf_counter() - t0
        n.pronline_reset(ctx)
        return (f"set {t_set:.2f} s, get {t_get:.3f} s, "
                f"sum={ranks.sum():.4f}")
    bench(f"pagerank_online RMAT-{SCALE}", BUILD_OUT_CSR, run_pronline)

if "konline" in algos:
    def run_konline(g):
        t0 = time.perf_counter()
        n.konline_set(ctx, g, ids, alpha=0.01, eps=1e-2)
        t_set = time.perf_counter() - t0
        iters = n.konline_iterations()
        n.konline_reset(ctx)
        return f"set {t_set:.2f} s, {iters} iterations"
    bench(f"katz_online RMAT-{SCALE} (alpha=0.01)",
          BUILD_IN_CSR | BUILD_OUT_CSR | BUILD_NO_PERM, run_konline)

if "louvain26" in algos:
    V26 = 1 << 26
    g = n.graph_rmat(ctx, 26, 16 * V26, seed=1,
                     flags=BUILD_SYM_CSR | BUILD_WEIGHTED)
    print(f"[louvain RMAT-26w] graph built ({n.graph_build_ms(g):.0f} ms)",
          flush=True)
    t0 = time.perf_counter()
    comm, nc = n.louvain(ctx, g, V26)
    t1 = time.perf_counter()
    print(f"[louvain RMAT-26w] {t1 - t0:.1f} s, {nc} communities", flush=True)
    n.graph_destroy(ctx, g)

n.destroy(ctx)
print("OK", flush=True)
