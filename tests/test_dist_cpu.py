"""Multi-process correctness of the sharded PageRank scheme, on CPU with the
gloo backend (world_size 2): each rank computes its owned destination rows
from the full contrib vector, all-gathers the owned rank/contrib slices, and
the assembled result must equal the oracle's single-process PageRank.

This is the same dataflow mgx_pagerank_start_dist runs on GPUs with RCCL
(memgraph_amd/csrc/pagerank.hip + comm.cpp); here the shard arithmetic and
gather pattern are validated where gloo runs."""
import os

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

from memgraph_amd.sharding import shard_range, shard_range_clamped, shard_size

WORLD = 2


def _numpy_pagerank_sharded(rank, world, nv, src, dst, iters, damping):
    outdeg = np.bincount(src, minlength=nv).astype(np.float64)
    inv = np.where(outdeg > 0, 1.0 / np.maximum(outdeg, 1), 0.0)
    b, e = shard_range(nv, world, rank)
    bc, ec = shard_range_clamped(nv, world, rank)
    own = (dst >= bc) & (dst < ec)
    src_l, dst_l = src[own], dst[own]
    s = shard_size(nv, world)

    base = (1.0 - damping) / nv
    padded = world * s
    rank_vec = np.full(padded, 0.0)
    rank_vec[:nv] = 1.0 / nv
    contrib = np.zeros(padded)
    contrib[:nv] = rank_vec[:nv] * inv
    for _ in range(iters):
        acc = np.zeros(s)
        np.add.at(acc, dst_l - b, contrib[src_l])
        new_slice = base + damping * acc
        if ec < e:  # padded tail of the last rank
            new_slice[ec - b:] = 0.0
        # allgather the owned slices (the ncclAllGather analogue)
        t = torch.from_numpy(new_slice)
        out = [torch.zeros(s, dtype=torch.float64) for _ in range(world)]
        dist.all_gather(out, t)
        rank_vec = torch.cat(out).numpy()
        contrib = np.zeros(padded)
        contrib[:nv] = rank_vec[:nv] * inv
    r = rank_vec[:nv]
    return r / r.sum()


def _worker(rank, world, port, nv, src, dst, expected):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    try:
        got = _numpy_pagerank_sharded(rank, world, nv, src, dst, iters=20, damping=0.85)
        assert np.abs(got - expected).max() < 1e-12, np.abs(got - expected).max()
    finally:
        dist.destroy_process_group()


def test_sharded_pagerank_matches_oracle(oracle):
    src, dst = oracle.gen_rmat(12, 16 * (1 << 12), seed=1)
    expected, _ = oracle.pagerank(1 << 12, src, dst, max_iterations=20, eps=0.0)
    mp.spawn(_worker, args=(WORLD, 29511, 1 << 12, src, dst, expected), nprocs=WORLD,
             join=True)


def test_shard_ranges_cover_and_pad():
    for nv in [1, 7, 64, 100, 101]:
        for world in [1, 2, 4, 8]:
            s = shard_size(nv, world)
            assert s * world >= nv
            covered = []
            for r in range(world):
                b, e = shard_range(nv, world, r)
                assert e - b == s
                bc, ec = shard_range_clamped(nv, world, r)
                covered.extend(range(bc, ec))
            assert covered == list(range(nv))


@pytest.mark.parametrize("world", [2, 4])
def test_uneven_tail_rank(world):
    nv = 10
    total = 0
    for r in range(world):
        bc, ec = shard_range_clamped(nv, world, r)
        total += ec - bc
    assert total == nv
