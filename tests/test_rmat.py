"""The three generator implementations (numpy / oracle C++ / HIP device)
must agree bit-for-bit; CPU side checks numpy vs C++ here, the HIP side is
checked in test_gpu_graph.py."""
import numpy as np

from memgraph_amd import rmat


def test_rmat_matches_oracle(oracle):
    s_np, d_np = rmat.gen_rmat(14, 20000, seed=1)
    s_c, d_c = oracle.gen_rmat(14, 20000, seed=1)
    assert np.array_equal(s_np, s_c)
    assert np.array_equal(d_np, d_c)


def test_uniform_matches_oracle(oracle):
    s_np, d_np = rmat.gen_uniform(10000, 50000, seed=42)
    s_c, d_c = oracle.gen_uniform(10000, 50000, seed=42)
    assert np.array_equal(s_np, s_c)
    assert np.array_equal(d_np, d_c)


def test_weights_match_oracle(oracle):
    w_np = rmat.gen_weights(30000, seed=7)
    w_c = oracle.gen_weights(30000, seed=7)
    assert np.array_equal(w_np, w_c)
    assert w_np.min() >= 0.0 and w_np.max() < 1.0


def test_rmat_determinism_and_range():
    s1, d1 = rmat.gen_rmat(10, 1000, seed=5)
    s2, d2 = rmat.gen_rmat(10, 1000, seed=5)
    assert np.array_equal(s1, s2) and np.array_equal(d1, d2)
    assert s1.min() >= 0 and s1.max() < 1024
    assert d1.min() >= 0 and d1.max() < 1024
    s3, _ = rmat.gen_rmat(10, 1000, seed=6)
    assert not np.array_equal(s1, s3)


def test_rmat_skew():
    # Graph500 parameters produce a skewed degree distribution.
    s, _ = rmat.gen_rmat(16, 16 * (1 << 16), seed=1)
    deg = np.bincount(s, minlength=1 << 16)
    assert deg.max() > 50 * deg.mean()
