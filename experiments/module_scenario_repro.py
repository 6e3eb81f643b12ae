import faulthandler; faulthandler.enable()

import sys, ctypes, os
sys.path.insert(0, '/root/repo/tests/mock')
from harness import ModuleHost
os.environ["MGX_PRONLINE_SEED"] = "99"
h = ModuleHost("pagerank_online")
h.mock.mock_result_string.restype = ctypes.c_char_p
# e2e graph
nodes = [0, 1, 2, 3, 4, 5]
src = [0, 1, 2, 3, 3, 3]
dst = [1, 2, 0, 3, 4, 5]
h.load_graph(nodes, [nodes.index(s) for s in src], [nodes.index(d) for d in dst])
rows = h.call("set")
ranks = [h.row_double(i, "rank") for i in rows]
assert len(rows) == 6, rows
assert all(0 < r < 1 for r in ranks), ranks
rows = h.call("get")
assert len(rows) == 6
# grow the graph (4->6) and update through the module path
h.mock.mock_reset_graph()
for n in nodes + [6]:
    h.mock.mock_add_vertex(ctypes.c_int64(n))
for s, d in zip(src + [4], dst + [6]):
    h.mock.mock_add_edge(ctypes.c_int64(s), ctypes.c_int64(d))
# get on changed graph -> inconsistency error
try:
    h.call("get")
    raise SystemExit("expected inconsistency error")
except RuntimeError as e:
    assert "incosistent" in str(e), e
h.override_arg_node_list(0, [6])
h.override_arg_edge_list(1, [4], [6])
rows = h.call("update")
assert len(rows) == 7, rows
rows = h.call("reset")
msg = h.mock.mock_result_string(0, b"message").decode()
assert "Pagerank context is reset" in msg, msg
print("MODULE_OK")
