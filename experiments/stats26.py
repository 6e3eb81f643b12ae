import sys
import numpy as np
sys.path.insert(0, "/root/repo")
from memgraph_amd import rmat

scale = 26
V = 1 << scale
E = 16 * V
CH = 1 << 27
outdeg = np.zeros(V, dtype=np.int64)
ms = rmat.seed_mix(1)
# chunked src-only generation: replicate gen_rmat's bit loop per chunk
t_a, t_ab, t_abc = rmat.rmat_thresholds()
for off in range(0, E, CH):
    n = min(CH, E - off)
    idx = (np.arange(off, off + n, dtype=np.uint64)) * np.uint64(scale)
    s = np.zeros(n, dtype=np.uint64)
    with np.errstate(over="ignore"):
        for level in range(scale):
            h = rmat.hash64(ms, idx + np.uint64(level))
            row_bit = (h >= t_ab).astype(np.uint64)
            s = (s << np.uint64(1)) | row_bit
    outdeg += np.bincount(s.astype(np.int64), minlength=V)
    print("chunk", off // CH, flush=True)

nnz = int((outdeg > 0).sum())
order = np.argsort(-outdeg, kind="stable")
sd = outdeg[order]
cum = np.cumsum(sd)
print(f"scale=26 V={V} E={E} sources_with_outdeg>0={nnz} ({nnz*4/1e6:.0f} MB of live contrib)")
for mb in [2, 4, 8, 16, 32, 64, 128, 192, 256]:
    k = min(mb * (1 << 20) // 4, V)
    print(f"gathers from first {mb} MB: {cum[k-1]/E:.4f}")
np.save("/root/repo/experiments/outdeg26.npy", outdeg.astype(np.int32))
