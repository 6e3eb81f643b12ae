# ORACLE — TEST INFRASTRUCTURE ONLY (see oracle/oracle.h header).
# ctypes bindings for oracle/liboracle.so (restatement) and oracle/_ref/libref.so
# (the reference's own cores). Importable only from tests/, bench.py's
# cpu_baseline leg and __graft_entry__.smoke(); the product path never
# imports this package.
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))

_I64 = ctypes.POINTER(ctypes.c_int64)
_F64 = ctypes.POINTER(ctypes.c_double)


def _as_i64(a):
    a = np.ascontiguousarray(a, dtype=np.int64)
    return a, a.ctypes.data_as(_I64)


def _as_f64(a):
    a = np.ascontiguousarray(a, dtype=np.float64)
    return a, a.ctypes.data_as(_F64)


class _Lib:
    def __init__(self, path):
        self.lib = ctypes.CDLL(path)

    def _fn(self, name, restype):
        f = getattr(self.lib, name)
        f.restype = restype
        return f


class Oracle(_Lib):
    """Restatement library (oracle/liboracle.so)."""

    def __init__(self, path=None):
        super().__init__(path or os.path.join(_DIR, "liboracle.so"))

    def pagerank(self, n_vertices, src, dst, max_iterations=100, damping=0.85, eps=1e-5):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        iters = self._fn("oracle_pagerank", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_int64(max_iterations), ctypes.c_double(damping), ctypes.c_double(eps),
            out.ctypes.data_as(_F64))
        assert iters >= 0, "oracle_pagerank failed"
        return out, iters

    def wcc(self, n_vertices, src, dst):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.int64)
        n = self._fn("oracle_wcc", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            out.ctypes.data_as(_I64))
        assert n >= 0, "oracle_wcc failed"
        return out, n

    def katz(self, n_vertices, src, dst, alpha=0.2, epsilon=1e-2):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        iters = self._fn("oracle_katz", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_double(alpha), ctypes.c_double(epsilon), out.ctypes.data_as(_F64))
        assert iters >= 0, "oracle_katz failed"
        return out, iters

    def louvain(self, n_vertices, src, dst, weights=None, threshold=1e-6):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        pw = None
        if weights is not None:
            weights, pw = _as_f64(weights)
        out = np.zeros(n_vertices, dtype=np.int64)
        n = self._fn("oracle_louvain", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst, pw,
            ctypes.c_double(threshold), out.ctypes.data_as(_I64))
        assert n >= 0, "oracle_louvain failed"
        return out, n

    def modularity(self, n_vertices, src, dst, community, weights=None):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        community, pc = _as_i64(community)
        pw = None
        if weights is not None:
            weights, pw = _as_f64(weights)
        return self._fn("oracle_modularity", ctypes.c_double)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst, pw, pc)

    def betweenness(self, n_vertices, src, dst, directed=True, normalize=True):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        rc = self._fn("oracle_betweenness", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_int32(1 if directed else 0), ctypes.c_int32(1 if normalize else 0),
            out.ctypes.data_as(_F64))
        assert rc == 0, "oracle_betweenness failed"
        return out

    def gen_rmat(self, scale, n_edges, seed=1, a=0.57, b=0.19, c=0.19):
        src = np.zeros(n_edges, dtype=np.int64)
        dst = np.zeros(n_edges, dtype=np.int64)
        self._fn("oracle_gen_rmat", None)(
            ctypes.c_int64(scale), ctypes.c_int64(n_edges), ctypes.c_uint64(seed),
            ctypes.c_double(a), ctypes.c_double(b), ctypes.c_double(c),
            src.ctypes.data_as(_I64), dst.ctypes.data_as(_I64))
        return src, dst

    def gen_uniform(self, n_vertices, n_edges, seed=42):
        src = np.zeros(n_edges, dtype=np.int64)
        dst = np.zeros(n_edges, dtype=np.int64)
        self._fn("oracle_gen_uniform", None)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(n_edges), ctypes.c_uint64(seed),
            src.ctypes.data_as(_I64), dst.ctypes.data_as(_I64))
        return src, dst

    def gen_weights(self, n_edges, seed=7):
        out = np.zeros(n_edges, dtype=np.float64)
        self._fn("oracle_gen_weights", None)(
            ctypes.c_int64(n_edges), ctypes.c_uint64(seed), out.ctypes.data_as(_F64))
        return out

    def pagerank_timed(self, n_vertices, src, dst, iterations, damping=0.85, n_threads=1):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        secs = self._fn("oracle_pagerank_timed", ctypes.c_double)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_int64(iterations), ctypes.c_double(damping), ctypes.c_int64(n_threads),
            out.ctypes.data_as(_F64))
        return out, secs


class Reference(_Lib):
    """The reference's own cores (oracle/_ref/libref.so). May be absent when
    oracle/ref/Makefile was never run (e.g. fresh checkout without build)."""

    def __init__(self, path=None):
        super().__init__(path or os.path.join(_DIR, "_ref", "libref.so"))

    def pagerank(self, n_vertices, src, dst, max_iterations=100, damping=0.85, eps=1e-5,
                 n_threads=1):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        n = self._fn("ref_pagerank", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_int64(max_iterations), ctypes.c_double(damping), ctypes.c_double(eps),
            ctypes.c_int64(n_threads), out.ctypes.data_as(_F64))
        assert n == n_vertices
        return out

    def pagerank_timed(self, n_vertices, src, dst, iterations, damping=0.85, n_threads=1):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        secs = self._fn("ref_pagerank_timed", ctypes.c_double)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_int64(iterations), ctypes.c_double(damping), ctypes.c_int64(n_threads),
            out.ctypes.data_as(_F64))
        return out, secs

    def betweenness(self, n_vertices, src, dst, directed=True, normalize=True,
                    n_threads=1):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        rc = self._fn("ref_betweenness", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_int32(1 if directed else 0), ctypes.c_int32(1 if normalize else 0),
            ctypes.c_int64(n_threads), out.ctypes.data_as(_F64))
        assert rc == 0
        return out

    def katz(self, n_vertices, src, dst, alpha=0.2, epsilon=1e-2):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        out = np.zeros(n_vertices, dtype=np.float64)
        n = self._fn("ref_katz", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst,
            ctypes.c_double(alpha), ctypes.c_double(epsilon), out.ctypes.data_as(_F64))
        assert n == n_vertices
        return out

    def louvain(self, n_vertices, src, dst, weights=None, threshold=1e-6, n_threads=1):
        src, psrc = _as_i64(src)
        dst, pdst = _as_i64(dst)
        pw = None
        if weights is not None:
            weights, pw = _as_f64(weights)
        out = np.zeros(n_vertices, dtype=np.int64)
        self._fn("ref_louvain", ctypes.c_int64)(
            ctypes.c_int64(n_vertices), ctypes.c_int64(len(src)), psrc, pdst, pw,
            ctypes.c_double(threshold), ctypes.c_int64(n_threads), out.ctypes.data_as(_I64))
        return out
