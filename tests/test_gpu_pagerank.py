"""GPU parity tests: HIP PageRank vs the fp64 oracle (which is itself pinned
bit-exact to the reference core — tests/test_oracle.py).

Parity bar: |r_gpu - r_cpu|inf <= 1e-6 after equal
iterations, post sum-normalize. The tolerance is written here, in the test.
"""
import json
import os

import numpy as np
import pytest

from memgraph_amd.native import BUILD_IN_CSR, Native

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(__file__), "golden")
TOL = 1e-6


@pytest.fixture(scope="module")
def nat():
    n = Native()
    if n.device_count() == 0:
        pytest.fail("gpu test run but no HIP device visible")
    return n


@pytest.fixture(scope="module")
def ctx(nat):
    c = nat.init(0)
    yield c
    nat.destroy(c)


def _load(name):
    with open(os.path.join(GOLDEN, name)) as f:
        return json.load(f)


def gpu_pagerank(nat, ctx, nv, src, dst, **kw):
    g = nat.graph_from_coo(ctx, src, dst, nv, flags=BUILD_IN_CSR)
    try:
        return nat.pagerank(ctx, g, nv, **kw)
    finally:
        nat.graph_destroy(ctx, g)


def test_unit_vectors(nat, ctx, oracle):
    for fx in _load("pagerank_unit.json"):
        nv = fx["n_vertices"]
        if nv == 0:
            continue
        rank, stats = gpu_pagerank(nat, ctx, nv, fx["src"], fx["dst"])
        exp, iters = oracle.pagerank(nv, fx["src"], fx["dst"])
        assert stats.iterations == iters, fx["name"]
        assert np.abs(rank - exp).max() <= TOL, fx["name"]
        # and against the published reference vectors at their tolerance
        pub = np.array(fx["expected_rank"])
        assert np.abs(rank - pub).max() < 1e-3, fx["name"]


def test_e2e_cases(nat, ctx, oracle):
    for fx in _load("e2e_cases.json"):
        if fx["algo"] != "pagerank" or fx["n_vertices"] == 0:
            continue
        max_iter = fx["args"][0] if fx["args"] else 100
        rank, stats = gpu_pagerank(nat, ctx, fx["n_vertices"], fx["src"], fx["dst"],
                                   max_iterations=max_iter)
        exp, iters = oracle.pagerank(fx["n_vertices"], fx["src"], fx["dst"],
                                     max_iterations=max_iter)
        assert stats.iterations == iters, fx["name"]
        assert np.abs(rank - exp).max() <= TOL, fx["name"]


def test_uniform_10k_50k(nat, ctx, oracle):
    # BASELINE.md config 1 shape.
    src, dst = oracle.gen_uniform(10000, 50000, seed=42)
    rank, stats = gpu_pagerank(nat, ctx, 10000, src, dst)
    exp, iters = oracle.pagerank(10000, src, dst)
    assert stats.iterations == iters
    assert np.abs(rank - exp).max() <= TOL
    assert abs(rank.sum() - 1.0) < 1e-9


def test_rmat18_20iters_fixed(nat, ctx, oracle):
    # Small-scale rehearsal of BASELINE.md config 2: fixed 20 iterations,
    # eps=0, RMAT via the same deterministic stream on both sides.
    scale, E = 18, 16 * (1 << 18)
    src, dst = oracle.gen_rmat(scale, E, seed=1)
    g_dev = Native().graph_rmat(ctx, scale, E, seed=1, flags=BUILD_IN_CSR)
    nat_local = Native()
    try:
        rank, stats = nat_local.pagerank(ctx, g_dev, 1 << scale, max_iterations=20, eps=0.0)
    finally:
        nat_local.graph_destroy(ctx, g_dev)
    exp, iters = oracle.pagerank(1 << scale, src, dst, max_iterations=20, eps=0.0)
    assert iters == 20 and stats.iterations == 20
    assert np.abs(rank - exp).max() <= TOL
    assert abs(rank.sum() - 1.0) < 1e-9


def test_multi_edges_self_loops_dangling(nat, ctx, oracle):
    # Edge cases the reference tests: multi-edges, self-loops, dangling
    # vertices, disconnected graphs (pagerank_test.cpp cases 3-5 shapes).
    rng = np.random.default_rng(7)
    for _ in range(5):
        nv = int(rng.integers(2, 500))
        ne = int(rng.integers(0, 3000))
        src = rng.integers(0, nv, ne)
        dst = rng.integers(0, nv, ne)
        rank, stats = gpu_pagerank(nat, ctx, nv, src, dst)
        exp, iters = oracle.pagerank(nv, src, dst)
        assert stats.iterations == iters
        assert np.abs(rank - exp).max() <= TOL


def test_zero_iterations(nat, ctx):
    rank, stats = gpu_pagerank(nat, ctx, 3, [0], [1], max_iterations=0)
    assert stats.iterations == 0
    assert np.allclose(rank, 1.0 / 3)


def test_run_api_matches_oneshot(nat, ctx, oracle):
    src, dst = oracle.gen_uniform(5000, 40000, seed=9)
    g = nat.graph_from_coo(ctx, src, dst, 5000, flags=BUILD_IN_CSR)
    try:
        run = nat.pagerank_start(ctx, g)
        nat.pagerank_iterate(run, 20)
        nat.sync(ctx)
        ms, n = nat.pagerank_timing(run)
        assert n == 20 and ms > 0
        rank = nat.pagerank_finish(run, 5000)
        one, _ = nat.pagerank(ctx, g, 5000, max_iterations=20, eps=0.0)
    finally:
        nat.graph_destroy(ctx, g)
    assert np.array_equal(rank, one)


def test_full_size_properties(nat, ctx):
    # Size-independent properties at a larger size (fits any MI355X easily):
    # probability distribution, positivity, deterministic repetition.
    scale, E = 22, 16 * (1 << 22)
    g = nat.graph_rmat(ctx, scale, E, seed=1, flags=BUILD_IN_CSR)
    try:
        r1, s1 = nat.pagerank(ctx, g, 1 << scale, max_iterations=20, eps=0.0)
    finally:
        nat.graph_destroy(ctx, g)
    assert abs(r1.sum() - 1.0) < 1e-6
    assert (r1 >= 0).all()
    assert s1.iterations == 20
    assert s1.sweep_launches == 20


def test_scan_order_invariance(nat, ctx, oracle):
    # The hot-first permutation is internal: results must be identical (to
    # fp tolerance) to the oracle regardless of input edge order.
    src, dst = oracle.gen_uniform(3000, 30000, seed=3)
    perm = np.random.default_rng(0).permutation(len(src))
    rank_a, _ = gpu_pagerank(nat, ctx, 3000, src, dst, max_iterations=15, eps=0.0)
    rank_b, _ = gpu_pagerank(nat, ctx, 3000, src[perm], dst[perm], max_iterations=15,
                             eps=0.0)
    exp, _ = oracle.pagerank(3000, src, dst, max_iterations=15, eps=0.0)
    assert np.abs(rank_a - exp).max() <= TOL
    assert np.abs(rank_b - exp).max() <= TOL


def test_stop_epsilon_respected(nat, ctx, oracle):
    # Default module args: eps=1e-5 must stop at the oracle's iteration.
    src, dst = oracle.gen_uniform(2000, 20000, seed=4)
    for eps in (1e-3, 1e-5, 1e-7):
        rank, stats = gpu_pagerank(nat, ctx, 2000, src, dst, max_iterations=500, eps=eps)
        exp, iters = oracle.pagerank(2000, src, dst, max_iterations=500, eps=eps)
        assert stats.iterations == iters, eps
        assert np.abs(rank - exp).max() <= TOL
