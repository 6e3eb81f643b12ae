"""Offline gather-locality statistics for the PageRank sweep (round-2 item 1).

For the hot-first (degree-descending) permuted in-CSR, quantify where the
measured 3.3x fabric over-fetch (32.6 GB vs 9.9 GB algorithmic at RMAT-26)
comes from: per-row distinct-line counts (spatial waste) vs cross-row reuse
distance (temporal, cache-capacity waste)."""
import sys
import numpy as np

sys.path.insert(0, "/root/repo")
from memgraph_amd import rmat

scale = int(sys.argv[1]) if len(sys.argv) > 1 else 22
ef = 16
V = 1 << scale
E = ef * V
src, dst = rmat.gen_rmat(scale, E, seed=1, dtype=np.int64)
src = src.astype(np.int64); dst = dst.astype(np.int64)

outdeg = np.bincount(src, minlength=V)
indeg = np.bincount(dst, minlength=V)

# duplicate edges
key = (dst.astype(np.uint64) << np.uint64(scale)) | src.astype(np.uint64)
uniq = np.unique(key).size
print(f"scale={scale} V={V} E={E} unique_edges={uniq} dup_frac={1-uniq/E:.4f}")
selfloops = int((src == dst).sum())
print(f"self_loops={selfloops}")

# hot-first permutation: order = argsort(-outdeg) stable
order = np.argsort(-outdeg, kind="stable")
perm = np.empty(V, dtype=np.int64); perm[order] = np.arange(V)
psrc = perm[src]; pdst = perm[dst]

# edge mass by permuted-source prefix (what fraction of gathers hit the hot prefix)
sorted_deg = outdeg[order]
cum = np.cumsum(sorted_deg)
for mb in [4, 32, 64, 128, 256]:
    k = min(mb * (1 << 20) // 4, V)   # prefix vertices fitting in mb MB of f32
    print(f"gathers from first {mb} MB of contrib: {cum[k-1]/E:.3f}")

# per-row distinct lines (16 sources/line) for the permuted sorted in-CSR
okey = (pdst.astype(np.uint64) << np.uint64(32)) | psrc.astype(np.uint64)
okey.sort()
prow = (okey >> np.uint64(32)).astype(np.int64)
pcol = (okey & np.uint64(0xFFFFFFFF)).astype(np.int64)
line = pcol >> 4
# distinct (row, line) pairs = line fetches if zero cross-row reuse
rl = (prow.astype(np.uint64) << np.uint64(32)) | line.astype(np.uint64)
new_rl = np.empty(E, dtype=bool); new_rl[0] = True
new_rl[1:] = rl[1:] != rl[:-1]
distinct_rl = int(new_rl.sum())
total_lines_touched = np.unique(line).size
print(f"edges={E} distinct(row,line)={distinct_rl} ({distinct_rl/E:.3f}/edge) "
      f"distinct lines={total_lines_touched}")
print(f"zero-temporal-reuse gather bytes/sweep = {distinct_rl*64/1e9:.2f} GB; "
      f"perfect-reuse floor = {total_lines_touched*64/1e9:.2f} GB; "
      f"algorithmic(8B/edge+20B/v) = {(E*8+V*20)/1e9:.2f} GB")
