"""katz_centrality_online oracle restatement, pinned EXACTLY against the
reference's own online katz core compiled from /root/reference
(oracle/_ref/libref_online.so) — the algorithm is deterministic
(DESIGN.md statistical-parity bar level 3 keeps the ordinary oracle bar)."""
import ctypes
import os

import numpy as np
import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

I64 = ctypes.c_int64
P64 = ctypes.POINTER(ctypes.c_int64)
PD = ctypes.POINTER(ctypes.c_double)


class KOracle:
    """ctypes driver for either liboracle (restatement) or libref_online."""

    def __init__(self, lib, prefix):
        self.lib = lib
        self.prefix = prefix
        set_fn = getattr(lib, prefix + "set")
        upd_fn = getattr(lib, prefix + "update")
        if prefix == "oracle_konline_":
            set_fn.argtypes = [I64, P64, I64, P64, P64, ctypes.c_double,
                               ctypes.c_double, PD]
            upd_fn.argtypes = [I64, P64, I64, P64, P64, P64, I64, P64, I64, P64,
                               P64, I64, P64, I64, PD]
        else:  # ref_katz_online_
            set_fn.argtypes = [I64, P64, I64, P64, P64, ctypes.c_double,
                               ctypes.c_double, PD]
            upd_fn.argtypes = [I64, P64, I64, P64, P64, P64, I64, P64, I64, P64,
                               P64, I64, P64, I64, PD]

    @staticmethod
    def arr(a):
        x = np.ascontiguousarray(a, dtype=np.int64)
        return x, x.ctypes.data_as(P64)

    def reset(self):
        getattr(self.lib, self.prefix + "reset")()

    def set(self, nodes, src, dst, alpha=0.2, eps=1e-2):
        nodes, pn = self.arr(nodes)
        src, ps = self.arr(src)
        dst, pd = self.arr(dst)
        out = np.zeros(len(nodes))
        getattr(self.lib, self.prefix + "set")(
            len(nodes), pn, len(src), ps, pd, ctypes.c_double(alpha),
            ctypes.c_double(eps), out.ctypes.data_as(PD))
        return out

    def update(self, nodes, src, dst, cv=(), ce=(), ce_idx=(), dv=(), de=()):
        nodes, pn = self.arr(nodes)
        src, ps = self.arr(src)
        dst, pd = self.arr(dst)
        cv, pcv = self.arr(cv)
        ce_f, pce = self.arr(np.asarray(ce, dtype=np.int64).reshape(-1))
        cei, pcei = self.arr(ce_idx)
        dv, pdv = self.arr(dv)
        de_f, pde = self.arr(np.asarray(de, dtype=np.int64).reshape(-1))
        out = np.zeros(len(nodes))
        getattr(self.lib, self.prefix + "update")(
            len(nodes), pn, len(src), ps, pd, pcv, I64(len(cv)), pce,
            I64(len(ce_f) // 2), pcei, pdv, I64(len(dv)), pde, I64(len(de_f) // 2),
            out.ctypes.data_as(PD))
        return out


@pytest.fixture
def both():
    ref_path = os.path.join(REPO, "oracle", "_ref", "libref_online.so")
    if not os.path.exists(ref_path):
        pytest.skip("_ref online lib not built (no /root/reference)")
    ours = KOracle(ctypes.CDLL(os.path.join(REPO, "oracle", "liboracle.so")),
                   "oracle_konline_")
    ref = KOracle(ctypes.CDLL(ref_path), "ref_katz_online_")
    ours.reset()
    ref.reset()
    yield ours, ref
    ours.reset()
    ref.reset()


# the reference e2e graph + variations
NODES = [0, 1, 2, 3, 4, 5]
SRC = [0, 1, 2, 3, 3, 3]
DST = [1, 2, 0, 3, 4, 5]


def assert_close(a, b, tol=1e-12):
    assert np.abs(a - b).max() <= tol, f"max diff {np.abs(a - b).max()}"


def test_konline_set(both):
    ours, ref = both
    a = ours.set(NODES, SRC, DST)
    b = ref.set(NODES, SRC, DST)
    assert_close(a, b)


def test_konline_set_uniform(both):
    ours, ref = both
    rng = np.random.RandomState(3)
    V, E = 40, 160
    src = rng.randint(0, V, E)
    dst = rng.randint(0, V, E)
    a = ours.set(list(range(V)), src, dst, alpha=0.1, eps=1e-2)
    b = ref.set(list(range(V)), src, dst, alpha=0.1, eps=1e-2)
    assert_close(a, b)


def test_konline_update_add_edge(both):
    ours, ref = both
    for o in both:
        o.set(NODES, SRC, DST)
    # add edge 4->5 (edge index = position in the new edge arrays)
    src2 = SRC + [4]
    dst2 = DST + [5]
    a = ours.update(NODES, src2, dst2, ce=[(4, 5)], ce_idx=[len(SRC)])
    b = ref.update(NODES, src2, dst2, ce=[(4, 5)], ce_idx=[len(SRC)])
    assert_close(a, b)


def test_konline_update_add_vertex(both):
    ours, ref = both
    for o in both:
        o.set(NODES, SRC, DST)
    nodes2 = NODES + [6]
    src2 = SRC + [4]
    dst2 = DST + [6]
    a = ours.update(nodes2, src2, dst2, cv=[6], ce=[(4, 6)], ce_idx=[len(SRC)])
    b = ref.update(nodes2, src2, dst2, cv=[6], ce=[(4, 6)], ce_idx=[len(SRC)])
    assert_close(a, b)


def test_konline_update_delete_edge(both):
    ours, ref = both
    for o in both:
        o.set(NODES, SRC, DST)
    # delete 3->4 (edge list without it)
    keep = [i for i in range(len(SRC)) if not (SRC[i] == 3 and DST[i] == 4)]
    src2 = [SRC[i] for i in keep]
    dst2 = [DST[i] for i in keep]
    a = ours.update(NODES, src2, dst2, de=[(3, 4)])
    b = ref.update(NODES, src2, dst2, de=[(3, 4)])
    assert_close(a, b)


def test_konline_update_delete_vertex(both):
    ours, ref = both
    for o in both:
        o.set(NODES, SRC, DST)
    # detach-delete node 5: edge 3->5 goes too
    nodes2 = [0, 1, 2, 3, 4]
    keep = [i for i in range(len(SRC)) if DST[i] != 5 and SRC[i] != 5]
    src2 = [SRC[i] for i in keep]
    dst2 = [DST[i] for i in keep]
    a = ours.update(nodes2, src2, dst2, dv=[5], de=[(3, 5)])
    b = ref.update(nodes2, src2, dst2, dv=[5], de=[(3, 5)])
    assert_close(a, b)


def test_konline_update_multi_edge(both):
    ours, ref = both
    src0 = SRC + [0]
    dst0 = DST + [1]  # parallel 0->1 exists from the start
    for o in both:
        o.set(NODES, src0, dst0)
    # add ANOTHER parallel 0->1
    src2 = src0 + [0]
    dst2 = dst0 + [1]
    a = ours.update(NODES, src2, dst2, ce=[(0, 1)], ce_idx=[len(src0)])
    b = ref.update(NODES, src2, dst2, ce=[(0, 1)], ce_idx=[len(src0)])
    assert_close(a, b)


def test_konline_sequential_updates(both):
    ours, ref = both
    rng = np.random.RandomState(7)
    V, E = 24, 72
    src = list(rng.randint(0, V, E))
    dst = list(rng.randint(0, V, E))
    for o in both:
        o.set(list(range(V)), src, dst, alpha=0.15, eps=1e-3)
    for step in range(3):
        s, d = int(rng.randint(0, V)), int(rng.randint(0, V))
        src2 = src + [s]
        dst2 = dst + [d]
        a = ours.update(list(range(V)), src2, dst2, ce=[(s, d)], ce_idx=[len(src)])
        b = ref.update(list(range(V)), src2, dst2, ce=[(s, d)], ce_idx=[len(src)])
        assert_close(a, b)
        src, dst = src2, dst2


def test_katz_online_module_registration():
    """katz_centrality_online.so registers the reference's exact procedures
    (katz_centrality_online_module.cpp:150-228)."""
    import subprocess
    import sys as _sys
    mockdir = os.path.join(os.path.dirname(__file__), "mock")
    modules = os.path.join(REPO, "memgraph_amd", "lib", "modules")
    if not os.path.exists(os.path.join(modules, "katz_centrality_online.so")):
        pytest.skip("katz_centrality_online.so not built")
    script = (
        "import sys, json\n"
        f"sys.path.insert(0, {mockdir!r})\n"
        "from harness import ModuleHost\n"
        "h = ModuleHost('katz_centrality_online')\n"
        "print(json.dumps(h.procedures()))\n")
    out = subprocess.run([_sys.executable, "-c", script], capture_output=True,
                         text=True)
    assert out.returncode == 0, out.stderr
    import json
    procs = json.loads(out.stdout)
    assert procs["set"]["args"] == [["alpha", "float"], ["epsilon", "float"]]
    assert procs["set"]["results"] == [["node", "node"], ["rank", "float"]]
    assert procs["get"]["args"] == []
    assert procs["update"]["args"] == [
        ["created_vertices", "nullable"], ["created_edges", "nullable"],
        ["deleted_vertices", "nullable"], ["deleted_edges", "nullable"]]
    assert procs["reset"]["results"] == [["message", "string"]]


def test_cd_online_module_registration():
    """community_detection_online.so registers the reference's exact
    procedures (community_detection_online_module.cpp:228-325)."""
    import subprocess
    import sys as _sys
    mockdir = os.path.join(os.path.dirname(__file__), "mock")
    modules = os.path.join(REPO, "memgraph_amd", "lib", "modules")
    if not os.path.exists(os.path.join(modules, "community_detection_online.so")):
        pytest.skip("community_detection_online.so not built")
    script = (
        "import sys, json\n"
        f"sys.path.insert(0, {mockdir!r})\n"
        "from harness import ModuleHost\n"
        "h = ModuleHost('community_detection_online')\n"
        "print(json.dumps(h.procedures()))\n")
    out = subprocess.run([_sys.executable, "-c", script], capture_output=True,
                         text=True)
    assert out.returncode == 0, out.stderr
    import json
    procs = json.loads(out.stdout)
    assert procs["set"]["args"] == [
        ["directed", "bool"], ["weighted", "bool"],
        ["similarity_threshold", "float"], ["exponent", "float"],
        ["min_value", "float"], ["weight_property", "string"],
        ["w_selfloop", "float"], ["max_iterations", "int"],
        ["max_updates", "int"]]
    assert procs["set"]["results"] == [["node", "node"], ["community_id", "int"]]
    assert procs["update"]["args"] == [
        ["createdVertices", "nullable"], ["createdEdges", "nullable"],
        ["updatedVertices", "nullable"], ["updatedEdges", "nullable"],
        ["deletedVertices", "nullable"], ["deletedEdges", "nullable"]]
    assert procs["reset"]["results"] == [["message", "string"]]
