"""Boundary tests (CPU): the four REAL drop-in module .so's load under the
reference's dlopen contract and register EXACTLY the reference's procedure
signatures. Execution needs a GPU (the modules fail loudly
without one — verified here too).
"""
import os
import subprocess
import sys

import pytest

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "mock"))
from harness import MODULES_DIR, ModuleHost  # noqa: E402

ALL_MODULES = ["pagerank", "katz_centrality", "community_detection",
               "weakly_connected_components", "betweenness_centrality"]


def require_built():
    for m in ALL_MODULES:
        if not os.path.exists(os.path.join(MODULES_DIR, m + ".so")):
            pytest.skip("module .so's not built (run make)")


# Each module loads in a subprocess (the mock registry is process-global and
# dlopen'd libs can't be unloaded reliably) and reports its registration.
CHECK_SCRIPT = r"""
import sys, json
sys.path.insert(0, {mockdir!r})
from harness import ModuleHost
h = ModuleHost({stem!r})
print(json.dumps(h.procedures()))
"""


def registration(stem):
    require_built()
    out = subprocess.run(
        [sys.executable, "-c",
         CHECK_SCRIPT.format(mockdir=os.path.join(os.path.dirname(__file__), "mock"),
                             stem=stem)],
        capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    import json
    return json.loads(out.stdout)


def test_pagerank_registration():
    procs = registration("pagerank")
    # pagerank_module.cpp:122-136
    assert procs["get"]["args"] == [
        ["max_iterations", "int"], ["damping_factor", "float"],
        ["stop_epsilon", "float"], ["num_of_threads", "int"]]
    assert procs["get"]["results"] == [["node", "node"], ["rank", "float"]]


def test_katz_registration():
    procs = registration("katz_centrality")
    # katz_centrality_module.cpp:57-74
    assert procs["get"]["args"] == [["alpha", "float"], ["epsilon", "float"]]
    assert procs["get"]["results"] == [["node", "node"], ["rank", "float"]]


def test_wcc_registration():
    procs = registration("weakly_connected_components")
    # connectivity_module.cpp:91-99
    assert procs["get"]["args"] == []
    assert procs["get"]["results"] == [["node", "node"], ["component_id", "int"]]


def test_community_registration():
    procs = registration("community_detection")
    # community_detection_module.cpp:111-136
    assert procs["get"]["args"] == [
        ["weight_property", "string"], ["coloring", "bool"],
        ["min_graph_shrink", "int"], ["community_alg_threshold", "float"],
        ["coloring_alg_threshold", "float"], ["num_of_threads", "int"]]
    assert procs["get"]["results"] == [["node", "node"], ["community_id", "int"]]
    # get_subgraph (community_detection_module.cpp:136-152)
    assert procs["get_subgraph"]["args"] == [
        ["subgraph_nodes", "list"], ["subgraph_relationships", "list"],
        ["weight_property", "string"], ["coloring", "bool"],
        ["min_graph_shrink", "int"], ["community_alg_threshold", "float"],
        ["coloring_alg_threshold", "float"], ["num_of_threads", "int"]]
    assert procs["get_subgraph"]["results"] == [["node", "node"], ["community_id", "int"]]


def test_betweenness_registration():
    procs = registration("betweenness_centrality")
    # betweenness_centrality_module.cpp:77-90
    assert procs["get"]["args"] == [
        ["directed", "bool"], ["normalized", "bool"], ["threads", "int"]]
    assert procs["get"]["results"] == [
        ["node", "node"], ["betweenness_centrality", "float"]]


FAIL_LOUD_SCRIPT = r"""
import sys
sys.path.insert(0, {mockdir!r})
from harness import ModuleHost
h = ModuleHost("pagerank")
h.load_graph([0, 1], [0], [1])
try:
    h.call("get")
    print("NO_ERROR")
except RuntimeError as e:
    print("ERR:" + str(e))
"""


def test_module_fails_loudly_without_gpu():
    """With no HIP device, calling the procedure must surface an error via
    mgp_result_set_error_msg — never silently fall back to CPU."""
    require_built()
    import ctypes
    native = os.path.join(os.path.dirname(MODULES_DIR), "libmgx_analytics.so")
    lib = ctypes.CDLL(native)
    if lib.mgx_device_count() > 0:
        pytest.skip("GPU present; fail-loud covered by gpu tests")
    out = subprocess.run(
        [sys.executable, "-c",
         FAIL_LOUD_SCRIPT.format(mockdir=os.path.join(os.path.dirname(__file__), "mock"))],
        capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    assert out.stdout.startswith("ERR:"), out.stdout
    assert "no HIP device" in out.stdout


COLORING_SCRIPT = r"""
import sys
sys.path.insert(0, {mockdir!r})
from harness import ModuleHost
h = ModuleHost("community_detection")
h.load_graph([0, 1], [0], [1])
h.override_arg(1, True)  # coloring=true
try:
    h.call("get")
    print("NO_ERROR")
except RuntimeError as e:
    print("ERR:" + str(e))
"""


def test_community_coloring_rejected():
    """coloring=true selects a different algorithm in the reference
    (runMultiPhaseColoring, louvain.cpp:42-48); the GPU backend rejects it
    explicitly instead of silently running the basic path. The check is
    before any device work, so it holds on CPU too."""
    require_built()
    out = subprocess.run(
        [sys.executable, "-c",
         COLORING_SCRIPT.format(mockdir=os.path.join(os.path.dirname(__file__), "mock"))],
        capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    assert out.stdout.startswith("ERR:"), out.stdout
    assert "coloring" in out.stdout
