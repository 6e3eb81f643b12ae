import sys, faulthandler
faulthandler.enable()
sys.path.insert(0, "/root/repo")
from memgraph_amd.native import BUILD_SYM_CSR, Native
n = Native(); print("devs", n.device_count(), flush=True)
ctx = n.init(0)
src = [0,1,2,3,4,5,2]; dst = [1,2,0,4,5,3,3]
g = n.graph_from_coo(ctx, src, dst, 6, flags=BUILD_SYM_CSR)
print("graph built", flush=True)
h, lv = n.leiden(ctx, g, 6, seed=3, cap=16)
print("leiden done", lv, h[:, :3].tolist(), flush=True)
n.graph_destroy(ctx, g); n.destroy(ctx)
