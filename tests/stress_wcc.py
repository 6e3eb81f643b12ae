"""Stress reproducer for the flaky WCC mismatch seen twice on GPU boxes
(test_wcc_random_bit_exact reporting n_components > n_vertices, which is
impossible with intact flag/scan buffers => suspected memory corruption).

Runs the exact seed-11 random-graph sequence of the failing test many times,
interleaved with pagerank builds to exercise the async-pool recycling that
differs between passing and failing runs, and prints full diagnostics on any
mismatch. Not a pytest test: invoked manually via gpurun.
"""
import sys

import numpy as np

sys.path.insert(0, ".")
from memgraph_amd.native import BUILD_IN_CSR, BUILD_SYM_CSR, Native  # noqa: E402
from oracle import Oracle  # noqa: E402


def main():
    reps = int(sys.argv[1]) if len(sys.argv) > 1 else 50
    nat = Native()
    orc = Oracle()
    ctx = nat.init(0)
    bad = 0
    for rep in range(reps):
        rng = np.random.default_rng(11)
        for it in range(6):
            nv = int(rng.integers(2, 2000))
            ne = int(rng.integers(0, 6000))
            src = rng.integers(0, nv, ne)
            dst = rng.integers(0, nv, ne)
            # mimic the failing runs' allocator traffic: a pagerank graph
            # built+destroyed before the wcc graph on odd reps
            if rep % 2 == 1:
                gp = nat.graph_from_coo(ctx, src, dst, nv, flags=BUILD_IN_CSR)
                nat.pagerank(ctx, gp, nv, max_iterations=3, eps=0.0)
                nat.graph_destroy(ctx, gp)
            g = nat.graph_from_coo(ctx, src, dst, nv, flags=BUILD_SYM_CSR)
            comp, n = nat.wcc(ctx, g, nv)
            nat.graph_destroy(ctx, g)
            exp, n_exp = orc.wcc(nv, src, dst)
            if n != n_exp or not np.array_equal(comp, exp):
                bad += 1
                print(f"MISMATCH rep={rep} it={it} nv={nv} ne={ne} "
                      f"n={n} n_exp={n_exp} comp_max={comp.max()} "
                      f"comp_min={comp.min()} "
                      f"n_distinct={len(np.unique(comp))} "
                      f"first_bad={np.flatnonzero(comp != exp)[:8].tolist()}",
                      flush=True)
        if rep % 10 == 9:
            print(f"rep {rep + 1}/{reps} done, {bad} mismatches", flush=True)
    nat.destroy(ctx)
    print(f"DONE reps={reps} mismatches={bad}")
    sys.exit(1 if bad else 0)


if __name__ == "__main__":
    main()
