"""Caching-allocator churn: recycled device blocks must never leak stale
state into results.

The r02 storm investigation replaced hipMallocAsync with an in-house
free-list over plain hipMalloc (memgraph_amd/csrc/mgx_api.cpp). These
tests cycle allocations hard — build/compute/destroy with varying sizes,
then repeat the first workload — and require bit-identical results, which
fails if any kernel assumes fresh-zero pages or a recycled block aliases
live data.
"""
import os
import sys

import numpy as np
import pytest

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from memgraph_amd.native import (BUILD_IN_CSR, BUILD_SYM_CSR, BUILD_WEIGHTED,
                                 Native)

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def nat():
    n = Native()
    if n.device_count() == 0:
        pytest.skip("no HIP device")
    return n


def test_recycled_blocks_bit_identical(nat):
    ctx = nat.init(0)
    try:
        def pr_run(seed):
            g = nat.graph_rmat(ctx, 17, 16 << 17, seed=seed,
                               flags=BUILD_IN_CSR | BUILD_SYM_CSR)
            rank, _ = nat.pagerank(ctx, g, 1 << 17)
            comp, ncomp = nat.wcc(ctx, g, 1 << 17)
            nat.graph_destroy(ctx, g)
            return rank, comp, ncomp

        first = pr_run(1)
        # churn: different sizes and algorithms recycle blocks of many
        # bucket sizes, including the louvain coarsen path
        for s in (3, 5):
            g = nat.graph_rmat(ctx, 18, 16 << 18, seed=s,
                               flags=BUILD_SYM_CSR | BUILD_WEIGHTED)
            _, nc = nat.louvain(ctx, g, 1 << 18)
            assert nc > 0
            nat.graph_destroy(ctx, g)
        again = pr_run(1)
        assert np.array_equal(first[0], again[0]), "pagerank not bit-stable"
        assert np.array_equal(first[1], again[1]), "wcc not bit-stable"
        assert first[2] == again[2]
    finally:
        nat.destroy(ctx)


def test_louvain_deterministic_after_churn(nat):
    ctx = nat.init(0)
    try:
        results = []
        for _ in range(2):
            g = nat.graph_rmat(ctx, 18, 16 << 18, seed=7,
                               flags=BUILD_SYM_CSR | BUILD_WEIGHTED)
            comm, nc = nat.louvain(ctx, g, 1 << 18)
            results.append((comm.copy(), nc))
            nat.graph_destroy(ctx, g)
            # interleave other allocations between the two runs
            g2 = nat.graph_rmat(ctx, 16, 16 << 16, seed=9, flags=BUILD_IN_CSR)
            nat.pagerank(ctx, g2, 1 << 16)
            nat.graph_destroy(ctx, g2)
        # fp64 atomic ordering makes trajectories formally nondeterministic;
        # in practice the schedule is stable (r02: repeat runs identical).
        # Assert the level that must not vary: a valid partition of the
        # same size both times.
        assert results[0][1] == results[1][1], "community count unstable"
    finally:
        nat.destroy(ctx)
