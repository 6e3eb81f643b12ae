"""ctypes bindings for libmgx_analytics.so (the gfx950 compute library).

Product-path plumbing for bench.py and the GPU tests. Fails loudly when the
library is missing or no HIP device is present — there is no CPU fallback.
"""
import ctypes
import os

import numpy as np

_DIR = os.path.dirname(os.path.abspath(__file__))
LIB_PATH = os.path.join(_DIR, "lib", "libmgx_analytics.so")

BUILD_IN_CSR = 1
BUILD_SYM_CSR = 2
BUILD_WEIGHTED = 4
BUILD_OUT_CSR = 8
BUILD_NO_PERM = 16

UNIQUE_ID_BYTES = 128

_I64 = ctypes.POINTER(ctypes.c_int64)
_F64 = ctypes.POINTER(ctypes.c_double)


class PagerankStats(ctypes.Structure):
    _fields_ = [
        ("iterations", ctypes.c_int64),
        ("iter_ms", ctypes.c_double),
        ("sweep_ms", ctypes.c_double),
        ("sweep_launches", ctypes.c_int64),
        ("csr_build_ms", ctypes.c_double),
        ("download_ms", ctypes.c_double),
    ]


class MgxError(RuntimeError):
    pass


class Native:
    def __init__(self, path=LIB_PATH):
        if not os.path.exists(path):
            raise MgxError(f"{path} not built — run `make` / __graft_entry__.build()")
        self.lib = ctypes.CDLL(path)
        self.lib.mgx_status_string.restype = ctypes.c_char_p
        self.lib.mgx_last_error.restype = ctypes.c_char_p
        self.lib.mgx_graph_build_ms.restype = ctypes.c_double
        self.lib.mgx_graph_num_vertices.restype = ctypes.c_int64
        self.lib.mgx_graph_num_edges.restype = ctypes.c_int64
        self.lib.mgx_graph_local_edges.restype = ctypes.c_int64

    def _check(self, status, what):
        if status != 0:
            msg = self.lib.mgx_status_string(status).decode()
            detail = self.lib.mgx_last_error().decode()
            raise MgxError(f"{what}: {msg} ({detail})")

    def device_count(self):
        return self.lib.mgx_device_count()

    def init(self, device=0):
        ctx = ctypes.c_void_p()
        self._check(self.lib.mgx_init(device, ctypes.byref(ctx)), "mgx_init")
        return ctx

    def destroy(self, ctx):
        self.lib.mgx_destroy(ctx)

    def sync(self, ctx):
        self._check(self.lib.mgx_sync(ctx), "mgx_sync")

    # --- graphs ---
    def graph_from_coo(self, ctx, src, dst, n_vertices, weights=None, flags=BUILD_IN_CSR):
        src = np.ascontiguousarray(src, dtype=np.int64)
        dst = np.ascontiguousarray(dst, dtype=np.int64)
        pw = None
        if weights is not None:
            weights = np.ascontiguousarray(weights, dtype=np.float64)
            pw = weights.ctypes.data_as(_F64)
        g = ctypes.c_void_p()
        self._check(
            self.lib.mgx_graph_from_coo(ctx, src.ctypes.data_as(_I64),
                                        dst.ctypes.data_as(_I64), pw,
                                        ctypes.c_int64(n_vertices),
                                        ctypes.c_int64(len(src)), ctypes.c_uint32(flags),
                                        ctypes.byref(g)), "mgx_graph_from_coo")
        return g

    def graph_rmat(self, ctx, scale, n_edges, seed=1, a=0.57, b=0.19, c=0.19,
                   flags=BUILD_IN_CSR, weight_seed=7):
        g = ctypes.c_void_p()
        self._check(
            self.lib.mgx_graph_rmat(ctx, ctypes.c_int(scale), ctypes.c_int64(n_edges),
                                    ctypes.c_uint64(seed), ctypes.c_double(a),
                                    ctypes.c_double(b), ctypes.c_double(c),
                                    ctypes.c_uint32(flags), ctypes.c_uint64(weight_seed),
                                    ctypes.byref(g)), "mgx_graph_rmat")
        return g

    def graph_uniform(self, ctx, n_vertices, n_edges, seed=42, flags=BUILD_IN_CSR,
                      weight_seed=7):
        g = ctypes.c_void_p()
        self._check(
            self.lib.mgx_graph_uniform(ctx, ctypes.c_int64(n_vertices),
                                       ctypes.c_int64(n_edges), ctypes.c_uint64(seed),
                                       ctypes.c_uint32(flags), ctypes.c_uint64(weight_seed),
                                       ctypes.byref(g)), "mgx_graph_uniform")
        return g

    def graph_rmat_sharded(self, ctx, scale, n_edges, row_begin, row_end, seed=1, a=0.57,
                           b=0.19, c=0.19):
        g = ctypes.c_void_p()
        self._check(
            self.lib.mgx_graph_rmat_sharded(ctx, ctypes.c_int(scale),
                                            ctypes.c_int64(n_edges), ctypes.c_uint64(seed),
                                            ctypes.c_double(a), ctypes.c_double(b),
                                            ctypes.c_double(c), ctypes.c_int64(row_begin),
                                            ctypes.c_int64(row_end), ctypes.byref(g)),
            "mgx_graph_rmat_sharded")
        return g

    def graph_destroy(self, ctx, g):
        self.lib.mgx_graph_destroy(ctx, g)

    def graph_build_ms(self, g):
        return self.lib.mgx_graph_build_ms(g)

    # --- algorithms ---
    def pagerank(self, ctx, g, n_vertices, max_iterations=100, damping=0.85, eps=1e-5):
        out = np.zeros(n_vertices, dtype=np.float64)
        stats = PagerankStats()
        self._check(
            self.lib.mgx_pagerank(ctx, g, ctypes.c_int64(max_iterations),
                                  ctypes.c_double(damping), ctypes.c_double(eps),
                                  out.ctypes.data_as(_F64), ctypes.byref(stats)),
            "mgx_pagerank")
        return out, stats

    def pagerank_start(self, ctx, g, damping=0.85):
        run = ctypes.c_void_p()
        self._check(self.lib.mgx_pagerank_start(ctx, g, ctypes.c_double(damping),
                                                ctypes.byref(run)), "mgx_pagerank_start")
        return run

    def pagerank_start_dist(self, ctx, g, row_begin, row_end, damping=0.85):
        run = ctypes.c_void_p()
        self._check(
            self.lib.mgx_pagerank_start_dist(ctx, g, ctypes.c_double(damping),
                                             ctypes.c_int64(row_begin),
                                             ctypes.c_int64(row_end), ctypes.byref(run)),
            "mgx_pagerank_start_dist")
        return run

    def pagerank_iterate(self, run, n):
        self._check(self.lib.mgx_pagerank_iterate(run, ctypes.c_int64(n)),
                    "mgx_pagerank_iterate")

    def pagerank_timing(self, run):
        ms = ctypes.c_double()
        n = ctypes.c_int64()
        self._check(self.lib.mgx_pagerank_timing(run, ctypes.byref(ms), ctypes.byref(n)),
                    "mgx_pagerank_timing")
        return ms.value, n.value

    def pagerank_finish(self, run, n_vertices=0, want_rank=True):
        out = np.zeros(n_vertices, dtype=np.float64) if want_rank else None
        p = out.ctypes.data_as(_F64) if want_rank and n_vertices else None
        self._check(self.lib.mgx_pagerank_finish(run, p), "mgx_pagerank_finish")
        return out

    def wcc(self, ctx, g, n_vertices):
        out = np.zeros(n_vertices, dtype=np.int64)
        n = ctypes.c_int64()
        self._check(self.lib.mgx_wcc(ctx, g, out.ctypes.data_as(_I64), ctypes.byref(n)),
                    "mgx_wcc")
        return out, n.value

    def katz(self, ctx, g, n_vertices, alpha=0.2, epsilon=1e-2):
        out = np.zeros(n_vertices, dtype=np.float64)
        iters = ctypes.c_int64()
        self._check(
            self.lib.mgx_katz(ctx, g, ctypes.c_double(alpha), ctypes.c_double(epsilon),
                              out.ctypes.data_as(_F64), ctypes.byref(iters)), "mgx_katz")
        return out, iters.value

    def louvain(self, ctx, g, n_vertices, threshold=1e-6):
        out = np.zeros(n_vertices, dtype=np.int64)
        n = ctypes.c_int64()
        self._check(
            self.lib.mgx_louvain(ctx, g, ctypes.c_double(threshold),
                                 out.ctypes.data_as(_I64), ctypes.byref(n)), "mgx_louvain")
        return out, n.value

    def betweenness(self, ctx, g, n_vertices, directed=True, normalize=True):
        out = np.zeros(n_vertices, dtype=np.float64)
        self._check(
            self.lib.mgx_betweenness(ctx, g, ctypes.c_int(1 if directed else 0),
                                     ctypes.c_int(1 if normalize else 0),
                                     out.ctypes.data_as(_F64)), "mgx_betweenness")
        return out

    # --- online pagerank (walk state is process-global, like the
    #     reference's context — algorithm_online/pagerank.cpp:49) ---
    def pronline_set(self, ctx, g, dense_to_mg, R=10, eps=0.2, seed=1):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        V = len(dense_to_mg)
        rank = np.zeros(V)
        self._check(
            self.lib.mgx_pronline_set(ctx, g, dense_to_mg.ctypes.data_as(_I64),
                                      ctypes.c_int64(R), ctypes.c_double(eps),
                                      ctypes.c_uint64(seed),
                                      rank.ctypes.data_as(_F64)),
            "mgx_pronline_set")
        return rank

    def pronline_get(self, ctx, dense_to_mg):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        V = len(dense_to_mg)
        rank = np.zeros(V)
        consistent = ctypes.c_int(0)
        self._check(
            self.lib.mgx_pronline_get(ctx, dense_to_mg.ctypes.data_as(_I64),
                                      ctypes.c_int64(V), rank.ctypes.data_as(_F64),
                                      ctypes.byref(consistent)),
            "mgx_pronline_get")
        return rank, consistent.value

    def pronline_update(self, ctx, g, dense_to_mg, cv=(), ce=(), dv=(), de=()):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        V = len(dense_to_mg)
        cv = np.ascontiguousarray(cv, dtype=np.int64)
        dv = np.ascontiguousarray(dv, dtype=np.int64)
        ce = np.ascontiguousarray(np.asarray(ce, dtype=np.int64).reshape(-1))
        de = np.ascontiguousarray(np.asarray(de, dtype=np.int64).reshape(-1))
        rank = np.zeros(V)
        self._check(
            self.lib.mgx_pronline_update(
                ctx, g, dense_to_mg.ctypes.data_as(_I64), cv.ctypes.data_as(_I64),
                ctypes.c_int64(len(cv)), ce.ctypes.data_as(_I64),
                ctypes.c_int64(len(ce) // 2), dv.ctypes.data_as(_I64),
                ctypes.c_int64(len(dv)), de.ctypes.data_as(_I64),
                ctypes.c_int64(len(de) // 2), rank.ctypes.data_as(_F64)),
            "mgx_pronline_update")
        return rank

    def pronline_reset(self, ctx):
        self._check(self.lib.mgx_pronline_reset(ctx), "mgx_pronline_reset")

    def pronline_stats(self, ctx):
        w = ctypes.c_int64(0)
        lw = ctypes.c_int64(0)
        le = ctypes.c_int64(0)
        self._check(self.lib.mgx_pronline_stats(ctx, ctypes.byref(w), ctypes.byref(lw),
                                                ctypes.byref(le)),
                    "mgx_pronline_stats")
        return w.value, lw.value, le.value

    # --- online katz (state is process-global, like the reference's
    #     context — query_modules katz.cpp:105) ---
    def konline_set(self, ctx, g, dense_to_mg, alpha=0.2, eps=1e-2):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        V = len(dense_to_mg)
        out = np.zeros(V)
        self._check(
            self.lib.mgx_konline_set(ctx, g, dense_to_mg.ctypes.data_as(_I64),
                                     ctypes.c_double(alpha), ctypes.c_double(eps),
                                     out.ctypes.data_as(_F64)),
            "mgx_konline_set")
        return out

    def konline_get(self, ctx, dense_to_mg):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        V = len(dense_to_mg)
        out = np.zeros(V)
        consistent = ctypes.c_int(0)
        self._check(
            self.lib.mgx_konline_get(ctx, dense_to_mg.ctypes.data_as(_I64),
                                     ctypes.c_int64(V), out.ctypes.data_as(_F64),
                                     ctypes.byref(consistent)),
            "mgx_konline_get")
        return out, consistent.value

    def konline_update(self, ctx, g, dense_to_mg, cv=(), ce=(), dv=(), de=()):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        cv = np.ascontiguousarray(cv, dtype=np.int64)
        dv = np.ascontiguousarray(dv, dtype=np.int64)
        ce = np.ascontiguousarray(np.asarray(ce, dtype=np.int64).reshape(-1))
        de = np.ascontiguousarray(np.asarray(de, dtype=np.int64).reshape(-1))
        out = np.zeros(len(dense_to_mg))
        self._check(
            self.lib.mgx_konline_update(
                ctx, g, dense_to_mg.ctypes.data_as(_I64), cv.ctypes.data_as(_I64),
                ctypes.c_int64(len(cv)), ce.ctypes.data_as(_I64),
                ctypes.c_int64(len(ce) // 2), dv.ctypes.data_as(_I64),
                ctypes.c_int64(len(dv)), de.ctypes.data_as(_I64),
                ctypes.c_int64(len(de) // 2), out.ctypes.data_as(_F64)),
            "mgx_konline_update")
        return out

    def konline_reset(self, ctx):
        self._check(self.lib.mgx_konline_reset(ctx), "mgx_konline_reset")

    def konline_iterations(self):
        self.lib.mgx_konline_iterations.restype = ctypes.c_int64
        return self.lib.mgx_konline_iterations()

    # --- LabelRankT online community detection (process-global state) ---
    def lrt_set(self, ctx, g, dense_to_mg, directed=False, weighted=False,
                similarity_threshold=0.7, exponent=4.0, min_value=0.1,
                w_selfloop=1.0, max_iterations=100, max_updates=5):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        out = np.full(len(dense_to_mg), -1, dtype=np.int64)
        self._check(
            self.lib.mgx_lrt_set(ctx, g, dense_to_mg.ctypes.data_as(_I64),
                                 ctypes.c_int(1 if directed else 0),
                                 ctypes.c_int(1 if weighted else 0),
                                 ctypes.c_double(similarity_threshold),
                                 ctypes.c_double(exponent),
                                 ctypes.c_double(min_value),
                                 ctypes.c_double(w_selfloop),
                                 ctypes.c_int64(max_iterations),
                                 ctypes.c_int64(max_updates),
                                 out.ctypes.data_as(_I64)),
            "mgx_lrt_set")
        return out

    def lrt_get(self, ctx, g, dense_to_mg):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        out = np.full(len(dense_to_mg), -1, dtype=np.int64)
        ran = ctypes.c_int(0)
        self._check(
            self.lib.mgx_lrt_get(ctx, g, dense_to_mg.ctypes.data_as(_I64),
                                 out.ctypes.data_as(_I64), ctypes.byref(ran)),
            "mgx_lrt_get")
        return out, ran.value

    def lrt_update(self, ctx, g, dense_to_mg, mv=(), me=(), dv=(), de=()):
        dense_to_mg = np.ascontiguousarray(dense_to_mg, dtype=np.int64)
        mv = np.ascontiguousarray(mv, dtype=np.int64)
        dv = np.ascontiguousarray(dv, dtype=np.int64)
        me = np.ascontiguousarray(np.asarray(me, dtype=np.int64).reshape(-1))
        de = np.ascontiguousarray(np.asarray(de, dtype=np.int64).reshape(-1))
        out = np.full(len(dense_to_mg), -1, dtype=np.int64)
        self._check(
            self.lib.mgx_lrt_update(
                ctx, g, dense_to_mg.ctypes.data_as(_I64), mv.ctypes.data_as(_I64),
                ctypes.c_int64(len(mv)), me.ctypes.data_as(_I64),
                ctypes.c_int64(len(me) // 2), dv.ctypes.data_as(_I64),
                ctypes.c_int64(len(dv)), de.ctypes.data_as(_I64),
                ctypes.c_int64(len(de) // 2), out.ctypes.data_as(_I64)),
            "mgx_lrt_update")
        return out

    def lrt_reset(self, ctx):
        self._check(self.lib.mgx_lrt_reset(ctx), "mgx_lrt_reset")

    # --- leiden ---
    def leiden(self, ctx, g, n_vertices, gamma=1.0, theta=0.01, resolution=0.01,
               max_iterations=(1 << 62), seed=1, cap=64):
        hier = np.full(n_vertices * cap, -1, dtype=np.int64)
        levels = np.zeros(n_vertices, dtype=np.int64)
        self._check(
            self.lib.mgx_leiden(ctx, g, ctypes.c_double(gamma), ctypes.c_double(theta),
                                ctypes.c_double(resolution),
                                ctypes.c_int64(max_iterations), ctypes.c_uint64(seed),
                                ctypes.c_int64(cap), hier.ctypes.data_as(_I64),
                                levels.ctypes.data_as(_I64)),
            "mgx_leiden")
        return hier.reshape(n_vertices, cap), levels

    # --- comm ---
    def comm_unique_id(self):
        buf = (ctypes.c_char * UNIQUE_ID_BYTES)()
        self._check(self.lib.mgx_comm_unique_id(buf), "mgx_comm_unique_id")
        return bytes(buf)

    def comm_init(self, ctx, rank, world, id_bytes):
        buf = (ctypes.c_char * UNIQUE_ID_BYTES).from_buffer_copy(id_bytes)
        self._check(self.lib.mgx_comm_init(ctx, rank, world, buf), "mgx_comm_init")
