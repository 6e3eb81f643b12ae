"""TEST INFRASTRUCTURE — drives the REAL drop-in module .so's against the
mgp host mock (libmgp_mock.so), mirroring memgraphd's loader contract
(reference module.cpp:855-913: dlopen, resolve mgp_init_module, call it,
expect 0)."""
import ctypes
import os

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
MOCK_PATH = os.path.join(REPO, "tests", "mock", "libmgp_mock.so")
MODULES_DIR = os.path.join(REPO, "memgraph_amd", "lib", "modules")


class ModuleHost:
    """One mock host process state. The mock is process-global, so use one
    instance at a time per module."""

    def __init__(self, module_stem):
        # RTLD_GLOBAL first so the module .so resolves mgp_* against the mock.
        self.mock = ctypes.CDLL(MOCK_PATH, mode=ctypes.RTLD_GLOBAL)
        self.mock.mock_module.restype = ctypes.c_void_p
        self.mock.mock_memory.restype = ctypes.c_void_p
        self.mock.mock_proc_name.restype = ctypes.c_char_p
        self.mock.mock_proc_arg_name.restype = ctypes.c_char_p
        self.mock.mock_proc_arg_type.restype = ctypes.c_char_p
        self.mock.mock_proc_result_name.restype = ctypes.c_char_p
        self.mock.mock_proc_result_type.restype = ctypes.c_char_p
        self.mock.mock_result_error.restype = ctypes.c_char_p
        self.mock.mock_result_double.restype = ctypes.c_double
        self.mock.mock_result_int.restype = ctypes.c_int64
        self.mock.mock_result_count.restype = ctypes.c_int64
        self.mock.mock_proc_count.restype = ctypes.c_int64
        self.mock.mock_proc_arg_count.restype = ctypes.c_int64
        self.mock.mock_proc_result_count.restype = ctypes.c_int64

        self.mock.mock_reset()
        path = os.path.join(MODULES_DIR, module_stem + ".so")
        # RTLD_NOW: fail loudly on any unresolved mgp_* import, exactly as
        # the reference loader does (module.cpp:861).
        self.module = ctypes.CDLL(path, mode=os.RTLD_NOW)
        rc = self.module.mgp_init_module(
            ctypes.c_void_p(self.mock.mock_module()),
            ctypes.c_void_p(self.mock.mock_memory()))
        assert rc == 0, f"mgp_init_module({module_stem}) returned {rc}"

    def procedures(self):
        out = {}
        for i in range(self.mock.mock_proc_count()):
            name = self.mock.mock_proc_name(i).decode()
            nm = name.encode()
            args = []
            for j in range(self.mock.mock_proc_arg_count(nm)):
                args.append((self.mock.mock_proc_arg_name(nm, j).decode(),
                             self.mock.mock_proc_arg_type(nm, j).decode()))
            results = []
            for j in range(self.mock.mock_proc_result_count(nm)):
                results.append((self.mock.mock_proc_result_name(nm, j).decode(),
                                self.mock.mock_proc_result_type(nm, j).decode()))
            out[name] = {"args": args, "results": results}
        return out

    def load_graph(self, node_props, src, dst, weights=None, weight_prop="weight"):
        self.mock.mock_reset_graph()
        for pid in node_props:
            self.mock.mock_add_vertex(ctypes.c_int64(pid))
        for i in range(len(src)):
            a = ctypes.c_int64(int(node_props[src[i]]))
            b = ctypes.c_int64(int(node_props[dst[i]]))
            if weights is not None:
                self.mock.mock_add_edge_weighted(a, b, weight_prop.encode(),
                                                 ctypes.c_double(float(weights[i])))
            else:
                self.mock.mock_add_edge(a, b)

    def override_arg_node_list(self, pos, mg_ids):
        import numpy as np
        a = np.ascontiguousarray(mg_ids, dtype=np.int64)
        self.mock.mock_override_arg_node_list(
            ctypes.c_int64(pos), a.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            ctypes.c_int64(len(a)))

    def override_arg_edge_list(self, pos, from_ids, to_ids):
        import numpy as np
        f = np.ascontiguousarray(from_ids, dtype=np.int64)
        t = np.ascontiguousarray(to_ids, dtype=np.int64)
        self.mock.mock_override_arg_edge_list(
            ctypes.c_int64(pos), f.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)),
            t.ctypes.data_as(ctypes.POINTER(ctypes.c_int64)), ctypes.c_int64(len(f)))

    def override_arg(self, pos, value):
        if isinstance(value, bool):
            self.mock.mock_override_arg_bool(ctypes.c_int64(pos),
                                             ctypes.c_int64(1 if value else 0))
            return
        if isinstance(value, int):
            self.mock.mock_override_arg_int(ctypes.c_int64(pos), ctypes.c_int64(value))
        elif isinstance(value, float):
            self.mock.mock_override_arg_double(ctypes.c_int64(pos), ctypes.c_double(value))
        elif isinstance(value, str):
            self.mock.mock_override_arg_string(ctypes.c_int64(pos), value.encode())
        else:
            raise TypeError(type(value))

    def call(self, proc="get"):
        rc = self.mock.mock_call(proc.encode())
        if rc == 1:
            raise RuntimeError("procedure error: " + self.mock.mock_result_error().decode())
        assert rc == 0, f"procedure {proc} not found"
        rows = []
        for i in range(self.mock.mock_result_count()):
            rows.append(i)
        return rows

    def row_int(self, i, field):
        return self.mock.mock_result_int(ctypes.c_int64(i), field.encode())

    def row_double(self, i, field):
        return self.mock.mock_result_double(ctypes.c_int64(i), field.encode())
