"""GPU graph-construction tests: the device generator is bit-identical to
the numpy/C implementations, and device CSR build handles edge cases."""
import ctypes

import numpy as np
import pytest

from memgraph_amd import rmat
from memgraph_amd.native import BUILD_IN_CSR, BUILD_SYM_CSR, Native

pytestmark = pytest.mark.gpu

_I64 = ctypes.POINTER(ctypes.c_int64)


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


def gen_on_device(nat, ctx, is_rmat, scale, nv, ne, seed):
    src = np.zeros(ne, dtype=np.int64)
    dst = np.zeros(ne, dtype=np.int64)
    nat._check(
        nat.lib.mgx_gen_edges_to_host(ctx, ctypes.c_int(1 if is_rmat else 0),
                                      ctypes.c_int(scale), ctypes.c_int64(nv),
                                      ctypes.c_int64(ne), ctypes.c_uint64(seed),
                                      ctypes.c_double(0.57), ctypes.c_double(0.19),
                                      ctypes.c_double(0.19),
                                      src.ctypes.data_as(_I64), dst.ctypes.data_as(_I64)),
        "mgx_gen_edges_to_host")
    return src, dst


def test_device_rmat_bit_identical(nat, ctx):
    s_d, d_d = gen_on_device(nat, ctx, True, 16, 1 << 16, 100000, seed=1)
    s_h, d_h = rmat.gen_rmat(16, 100000, seed=1)
    assert np.array_equal(s_d, s_h)
    assert np.array_equal(d_d, d_h)


def test_device_uniform_bit_identical(nat, ctx):
    s_d, d_d = gen_on_device(nat, ctx, False, 0, 10000, 50000, seed=42)
    s_h, d_h = rmat.gen_uniform(10000, 50000, seed=42)
    assert np.array_equal(s_d, s_h)
    assert np.array_equal(d_d, d_h)


def test_build_empty_graph(nat, ctx):
    g = nat.graph_from_coo(ctx, [], [], 5, flags=BUILD_IN_CSR | BUILD_SYM_CSR)
    nat.graph_destroy(ctx, g)


def test_build_rejects_out_of_range(nat, ctx):
    from memgraph_amd.native import MgxError
    with pytest.raises(MgxError):
        nat.graph_from_coo(ctx, [0], [7], 5, flags=BUILD_IN_CSR)


def test_build_ms_reported(nat, ctx):
    g = nat.graph_rmat(ctx, 16, 100000, seed=1, flags=BUILD_IN_CSR)
    try:
        assert nat.graph_build_ms(g) > 0
    finally:
        nat.graph_destroy(ctx, g)
