"""GPU Louvain parity vs the oracle (pinned to grappolo 1-thread).

For integer-valued weights every fp64 sum in the sweep is EXACT, so the GPU
Jacobi sweep must take the same decision sequence as the sequential oracle
=> identical partitions (not just equivalent). With arbitrary fp weights,
atomic summation order can flip near-ties, so parity there is partition on
the golden cases + modularity within 1e-3 (BASELINE.md config 4 bar).
"""
import json
import os

import numpy as np
import pytest

from memgraph_amd.native import BUILD_SYM_CSR, BUILD_WEIGHTED, Native

pytestmark = pytest.mark.gpu

GOLDEN = os.path.join(os.path.dirname(__file__), "golden")


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


def gpu_louvain(nat, ctx, nv, src, dst, weights=None, threshold=1e-6):
    flags = BUILD_SYM_CSR | (BUILD_WEIGHTED if weights is not None else 0)
    g = nat.graph_from_coo(ctx, src, dst, nv, weights=weights, flags=flags)
    try:
        return nat.louvain(ctx, g, nv, threshold=threshold)
    finally:
        nat.graph_destroy(ctx, g)


def test_goldens_exact_partitions(nat, ctx, oracle):
    with open(os.path.join(GOLDEN, "e2e_cases.json")) as f:
        cases = json.load(f)
    for fx in cases:
        if fx["algo"] != "community_detection" or not fx["src"]:
            continue
        comm, n = gpu_louvain(nat, ctx, fx["n_vertices"], fx["src"], fx["dst"],
                              weights=fx["weights"])
        exp, n_exp = oracle.louvain(fx["n_vertices"], fx["src"], fx["dst"],
                                    weights=fx["weights"])
        assert np.array_equal(comm, exp), (fx["name"], comm, exp)
        assert n == n_exp, fx["name"]


def test_random_unweighted_exact(nat, ctx, oracle):
    rng = np.random.default_rng(17)
    for _ in range(6):
        nv = int(rng.integers(4, 1500))
        ne = int(rng.integers(1, 8000))
        src = rng.integers(0, nv, ne)
        dst = rng.integers(0, nv, ne)
        comm, n = gpu_louvain(nat, ctx, nv, src, dst)
        exp, n_exp = oracle.louvain(nv, src, dst)
        assert np.array_equal(comm, exp)
        assert n == n_exp


def test_rmat_unweighted_exact(nat, ctx, oracle):
    src, dst = oracle.gen_rmat(14, 8 * (1 << 14), seed=21)
    comm, n = gpu_louvain(nat, ctx, 1 << 14, src, dst)
    exp, n_exp = oracle.louvain(1 << 14, src, dst)
    assert np.array_equal(comm, exp)
    assert n == n_exp


def test_rmat16_unweighted_exact(nat, ctx, oracle):
    # Mid-scale exact parity (integer weights => every fp64 sum exact):
    # exercises multiple coarsening phases and both sweep paths.
    src, dst = oracle.gen_rmat(16, 8 * (1 << 16), seed=33)
    comm, n = gpu_louvain(nat, ctx, 1 << 16, src, dst)
    exp, n_exp = oracle.louvain(1 << 16, src, dst)
    assert np.array_equal(comm, exp)
    assert n == n_exp


def test_hub_rows_exact(nat, ctx, oracle):
    # Rows above the wave/block degree split (>= 256 neighbours) exercise
    # the global-pool path.
    nv = 2000
    src = [0] * 1200 + list(range(1, 400))
    dst = list(np.arange(1200) % (nv - 1) + 1) + [0] * 399
    comm, n = gpu_louvain(nat, ctx, nv, src, dst)
    exp, n_exp = oracle.louvain(nv, src, dst)
    assert np.array_equal(comm, exp)
    assert n == n_exp


def test_weighted_modularity_tolerance(nat, ctx, oracle):
    rng = np.random.default_rng(19)
    nv, ne = 3000, 20000
    src = rng.integers(0, nv, ne)
    dst = rng.integers(0, nv, ne)
    # f32-representable weights so the GPU quantization is lossless.
    w = rng.random(ne).astype(np.float32).astype(np.float64)
    comm, _ = gpu_louvain(nat, ctx, nv, src, dst, weights=w)
    exp, _ = oracle.louvain(nv, src, dst, weights=w)
    q_gpu = oracle.modularity(nv, src, dst, comm, weights=w)
    q_cpu = oracle.modularity(nv, src, dst, exp, weights=w)
    assert abs(q_gpu - q_cpu) <= 1e-3, (q_gpu, q_cpu)


def test_isolated_vertices_minus_one(nat, ctx, oracle):
    # Two triangles + 3 isolated vertices: isolated vertices end at -1 when
    # >= 3 sweeps run (the reference's rotation artifact).
    src = [0, 1, 2, 3, 4, 5]
    dst = [1, 2, 0, 4, 5, 3]
    nv = 9
    comm, n = gpu_louvain(nat, ctx, nv, src, dst)
    exp, n_exp = oracle.louvain(nv, src, dst)
    assert np.array_equal(comm, exp)
    assert n == n_exp


def test_no_edges(nat, ctx):
    comm, n = gpu_louvain(nat, ctx, 3, [], [])
    assert n == 0
    assert np.all(comm == -1)


def test_threshold_sweep_exact(nat, ctx, oracle):
    # Non-default thresholds (the module's `threshold` argument): the
    # phase-stop decision depends on the sweep loop returning the
    # reference's prevMod, not currMod — loose thresholds expose any
    # mismatch (tests/test_oracle_params.py pins the oracle side).
    rng = np.random.default_rng(23)
    for thr in (1e-1, 1e-2, 1e-4):
        for _ in range(3):
            nv = int(rng.integers(4, 1200))
            ne = int(rng.integers(1, 6000))
            src = rng.integers(0, nv, ne)
            dst = rng.integers(0, nv, ne)
            comm, n = gpu_louvain(nat, ctx, nv, src, dst, threshold=thr)
            exp, n_exp = oracle.louvain(nv, src, dst, threshold=thr)
            assert np.array_equal(comm, exp), thr
            assert n == n_exp, thr
