"""Reconcile csr_build_ms: build the RMAT-26 in-CSR
repeatedly in one process and print each build's device time."""
import sys
sys.path.insert(0, "/root/repo")
from memgraph_amd.native import BUILD_IN_CSR, Native
n = Native(); ctx = n.init(0)
scale = int(sys.argv[1]) if len(sys.argv) > 1 else 26
V = 1 << scale; E = 16 * V
for i in range(4):
    g = n.graph_rmat(ctx, scale, E, seed=1, flags=BUILD_IN_CSR)
    print(f"build {i}: {n.graph_build_ms(g):.1f} ms", flush=True)
    n.graph_destroy(ctx, g)
n.destroy(ctx)
