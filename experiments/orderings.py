"""Candidate vertex orderings for the PageRank in-CSR (experiment tooling).

Each ordering is a permutation new<-old; the CSR is rebuilt in the permuted
space exactly as graph_build.hip does (sort by (perm[dst]<<32)|perm[src]),
then written as row_ptr.bin/col.bin for sim.c.
"""
import os
import sys

import numpy as np

sys.path.insert(0, "/root/repo")
from memgraph_amd import rmat


def build_csr(psrc, pdst, V):
    key = (pdst.astype(np.uint64) << np.uint64(32)) | psrc.astype(np.uint64)
    key.sort()
    col = (key & np.uint64(0xFFFFFFFF)).astype(np.int32)
    row = (key >> np.uint64(32)).astype(np.int64)
    counts = np.bincount(row, minlength=V)
    row_ptr = np.zeros(V + 1, dtype=np.uint32)
    np.cumsum(counts, out=row_ptr[1:])
    return row_ptr, col


def order_deg_desc(src, dst, V):
    outdeg = np.bincount(src, minlength=V)
    return np.argsort(-outdeg, kind="stable")


def order_hot_then_natural(src, dst, V, hot_mb=4):
    """Top-K by degree (K = hot_mb MB of f32 contrib), then original-id order."""
    outdeg = np.bincount(src, minlength=V)
    k = hot_mb * (1 << 20) // 4
    hot = np.argsort(-outdeg, kind="stable")[:k]
    mask = np.ones(V, dtype=bool)
    mask[hot] = False
    tail = np.nonzero(mask)[0]  # ascending original id
    return np.concatenate([hot, tail])


def order_hot_then_mindst(src, dst, V, hot_mb=4):
    """Hot prefix by degree; tail sources ordered by the (permuted) id of the
    smallest destination that gathers them, so a destination row's tail
    gathers become contiguous runs; ties by degree desc then id."""
    outdeg = np.bincount(src, minlength=V)
    dd = order_deg_desc(src, dst, V)
    perm_d = np.empty(V, dtype=np.int64)
    perm_d[dd] = np.arange(V)
    # min (deg-desc-permuted) destination per source
    mindst = np.full(V, np.int64(1 << 62))
    np.minimum.at(mindst, src, perm_d[dst])
    k = hot_mb * (1 << 20) // 4
    hot = dd[:k]
    mask = np.ones(V, dtype=bool)
    mask[hot] = False
    tail = np.nonzero(mask)[0]
    # sort tail by (mindst asc, degree desc, id asc)
    keys = np.lexsort((tail, -outdeg[tail], mindst[tail]))
    return np.concatenate([hot, tail[keys]])


def order_degbucket_then_mindst(src, dst, V):
    """Degree-bucketed (log2) major order, min-dst within bucket."""
    outdeg = np.bincount(src, minlength=V)
    dd = order_deg_desc(src, dst, V)
    perm_d = np.empty(V, dtype=np.int64)
    perm_d[dd] = np.arange(V)
    mindst = np.full(V, np.int64(1 << 62))
    np.minimum.at(mindst, src, perm_d[dst])
    logdeg = np.zeros(V, dtype=np.int64)
    nz = outdeg > 0
    logdeg[nz] = np.floor(np.log2(outdeg[nz])).astype(np.int64) + 1
    ids = np.arange(V)
    keys = np.lexsort((ids, mindst, -logdeg))
    return ids[keys]


ORDERINGS = {
    "deg_desc": order_deg_desc,
    "hot4_natural": order_hot_then_natural,
    "hot4_mindst": order_hot_then_mindst,
    "degbucket_mindst": order_degbucket_then_mindst,
}


def main():
    scale = int(sys.argv[1]) if len(sys.argv) > 1 else 24
    which = sys.argv[2:] if len(sys.argv) > 2 else list(ORDERINGS)
    V = 1 << scale
    E = 16 * V
    src, dst = rmat.gen_rmat(scale, E, seed=1, dtype=np.int64)
    src = src.astype(np.int64)
    dst = dst.astype(np.int64)
    outdir = f"/tmp/simdata{scale}"
    os.makedirs(outdir, exist_ok=True)
    for name in which:
        order = ORDERINGS[name](src, dst, V).astype(np.int64)
        perm = np.empty(V, dtype=np.int64)
        perm[order] = np.arange(V)
        row_ptr, col = build_csr(perm[src], perm[dst], V)
        row_ptr.tofile(f"{outdir}/{name}.row_ptr.bin")
        col.tofile(f"{outdir}/{name}.col.bin")
        print(f"wrote {name} (scale {scale})", flush=True)


if __name__ == "__main__":
    main()
