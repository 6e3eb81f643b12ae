import sys, time, faulthandler
faulthandler.enable()
sys.path.insert(0, "/root/repo")
from memgraph_amd.native import BUILD_SYM_CSR, BUILD_WEIGHTED, Native
n = Native(path="/root/repo/build/asan/libmgx_asan.so")
ctx = n.init(0)
scale = int(sys.argv[1]) if len(sys.argv) > 1 else 24
V = 1 << scale; E = 16 * V
g = n.graph_rmat(ctx, scale, E, seed=1, flags=BUILD_SYM_CSR | BUILD_WEIGHTED)
print(f"graph built ({n.graph_build_ms(g):.0f} ms)", flush=True)
t0 = time.perf_counter()
comm, nc = n.louvain(ctx, g, V)
print(f"COMPLETED: {time.perf_counter()-t0:.1f} s, {nc} communities", flush=True)
