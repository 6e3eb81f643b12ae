"""Deterministic synthetic graph generators (numpy port).

Bit-identical to include/mgx_graphgen.h (used by the HIP device generator
and the oracle's C++ generator): counter-based splitmix64 streams, integer
quadrant thresholds — see that header for the contract. Tests verify the
three implementations agree.

RMAT parameters follow the reference generator's own defaults
(/root/reference/src/mage/cpp/cugraph_module/algorithms/graph_generator.cu:143-147:
a=0.57, b=0.19, c=0.19, clip_and_flip off; multi-edges/self-loops kept, as
pagerank.hpp:27 allows).
"""
import numpy as np

_U64 = np.uint64
_GOLDEN = _U64(0x9E3779B97F4A7C15)
_MIX1 = _U64(0xBF58476D1CE4E5B9)
_MIX2 = _U64(0x94D049BB133111EB)
_SEEDK = _U64(0x5851F42D4C957F2D)


def _mix64(x):
    x = x.astype(np.uint64, copy=True) if isinstance(x, np.ndarray) else _U64(x)
    with np.errstate(over="ignore"):
        x ^= x >> _U64(30)
        x *= _MIX1
        x ^= x >> _U64(27)
        x *= _MIX2
        x ^= x >> _U64(31)
    return x


def seed_mix(seed):
    return _mix64(_U64(seed) ^ _SEEDK)


def hash64(mixed_seed, idx):
    """idx: uint64 ndarray -> uint64 ndarray."""
    with np.errstate(over="ignore"):
        return _mix64(_U64(mixed_seed) + idx * _GOLDEN)


def rmat_thresholds(a=0.57, b=0.19, c=0.19):
    two64 = 18446744073709551616.0
    return _U64(a * two64), _U64((a + b) * two64), _U64((a + b + c) * two64)


def gen_rmat(scale, n_edges, seed=1, a=0.57, b=0.19, c=0.19, dtype=np.int64):
    """Edge list of RMAT(scale) with V = 2**scale; returns (src, dst)."""
    ms = seed_mix(seed)
    t_a, t_ab, t_abc = rmat_thresholds(a, b, c)
    src = np.zeros(n_edges, dtype=np.uint64)
    dst = np.zeros(n_edges, dtype=np.uint64)
    idx_base = np.arange(n_edges, dtype=np.uint64)
    with np.errstate(over="ignore"):
        idx_base = idx_base * _U64(scale)
        for level in range(scale):
            h = hash64(ms, idx_base + _U64(level))
            row_bit = (h >= t_ab).astype(np.uint64)
            col_bit = (((h >= t_a) & (h < t_ab)) | (h >= t_abc)).astype(np.uint64)
            src |= row_bit << _U64(level)
            dst |= col_bit << _U64(level)
    return src.astype(dtype), dst.astype(dtype)


def gen_uniform(n_vertices, n_edges, seed=42, dtype=np.int64):
    ms = seed_mix(seed)
    i = np.arange(n_edges, dtype=np.uint64)
    with np.errstate(over="ignore"):
        src = hash64(ms, _U64(2) * i) % _U64(n_vertices)
        dst = hash64(ms, _U64(2) * i + _U64(1)) % _U64(n_vertices)
    return src.astype(dtype), dst.astype(dtype)


def gen_weights(n_edges, seed=7):
    """Edge weights in [0, 1), float64, 53-bit mantissa."""
    ms = seed_mix(seed)
    i = np.arange(n_edges, dtype=np.uint64)
    h = hash64(ms, i) >> _U64(11)
    return h.astype(np.float64) * (1.0 / 9007199254740992.0)
