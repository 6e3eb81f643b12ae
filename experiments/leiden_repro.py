import sys, faulthandler
faulthandler.enable()
sys.path.insert(0, "/root/repo")
import numpy as np
from memgraph_amd.native import BUILD_SYM_CSR, Native
n = Native(); ctx = n.init(0)
rng = np.random.RandomState(9)
V, E = 120, 480
src = list(rng.randint(0, V, E)); dst = list(rng.randint(0, V, E))
g = n.graph_from_coo(ctx, src, dst, V, flags=BUILD_SYM_CSR)
h, lv = n.leiden(ctx, g, V, seed=7, cap=16)
print("levels dist:", np.bincount(lv), flush=True)
for k in range(int(lv.max())):
    ids = [h[v][k] for v in range(V) if lv[v] > k]
    print(f"level {k}: distinct={len(set(ids))} n={len(ids)}", flush=True)
n.graph_destroy(ctx, g); n.destroy(ctx)
