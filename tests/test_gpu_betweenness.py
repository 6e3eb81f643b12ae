"""GPU betweenness (exact Brandes) vs the oracle (pinned to the reference
core in tests/test_oracle.py). Per-source dependencies are deterministic on
GPU (pull form); only the cross-source fp64 accumulation is atomic, so
values match to ~1e-9 relative (the reference's own thread pool has the
same property)."""
import json
import os

import numpy as np
import pytest

from memgraph_amd.native import BUILD_OUT_CSR, BUILD_SYM_CSR, Native

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


def gpu_bc(nat, ctx, nv, src, dst, directed=True, normalize=True):
    flags = BUILD_OUT_CSR if directed else BUILD_SYM_CSR
    g = nat.graph_from_coo(ctx, src, dst, nv, flags=flags)
    try:
        return nat.betweenness(ctx, g, nv, directed=directed, normalize=normalize)
    finally:
        nat.graph_destroy(ctx, g)


def _close(a, b):
    scale = np.maximum(np.abs(b), 1.0)
    return (np.abs(a - b) / scale).max() < 1e-9


def test_goldens(nat, ctx, oracle):
    with open(os.path.join(GOLDEN, "e2e_cases.json")) as f:
        cases = json.load(f)
    for fx in cases:
        if fx["algo"] != "betweenness" or fx["n_vertices"] == 0:
            continue
        args = fx["args"]
        directed = bool(args[0]) if len(args) > 0 else True
        normalize = bool(args[1]) if len(args) > 1 else True
        got = gpu_bc(nat, ctx, fx["n_vertices"], fx["src"], fx["dst"], directed, normalize)
        exp = oracle.betweenness(fx["n_vertices"], fx["src"], fx["dst"], directed,
                                 normalize)
        assert _close(got, exp), fx["name"]


def test_random_graphs(nat, ctx, oracle):
    rng = np.random.default_rng(29)
    for _ in range(5):
        nv = int(rng.integers(2, 800))
        ne = int(rng.integers(0, 4000))
        src = rng.integers(0, nv, ne)
        dst = rng.integers(0, nv, ne)
        for directed in (True, False):
            got = gpu_bc(nat, ctx, nv, src, dst, directed, True)
            exp = oracle.betweenness(nv, src, dst, directed, True)
            assert _close(got, exp), (nv, ne, directed)


def test_rmat(nat, ctx, oracle):
    src, dst = oracle.gen_rmat(11, 8 * (1 << 11), seed=31)
    got = gpu_bc(nat, ctx, 1 << 11, src, dst, True, False)
    exp = oracle.betweenness(1 << 11, src, dst, True, False)
    assert _close(got, exp)


def test_module_end_to_end(oracle):
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), "mock"))
    from harness import ModuleHost
    h = ModuleHost("betweenness_centrality")
    src = [0, 1, 2, 3]
    dst = [1, 2, 3, 0]
    h.load_graph(list(range(5)), src, dst)
    h.override_arg(0, False)  # directed
    h.override_arg(1, True)   # normalized
    rows = h.call("get")
    got = {h.row_int(i, "node"): h.row_double(i, "betweenness_centrality") for i in rows}
    exp = oracle.betweenness(5, src, dst, directed=False, normalize=True)
    for dense in range(5):
        assert abs(got[dense] - exp[dense]) < 1e-9
