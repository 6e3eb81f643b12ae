#!/usr/bin/env python3
"""Generate committed golden fixtures from the reference's own test suite.

Run in the DEV container only (needs /root/reference); the JSON outputs are
committed so GPU-side tests never read /root/reference.

Sources:
  1. e2e YAML cases: /root/reference/tests/mage/e2e/{pagerank_test,
     weakly_connected_components_test,community_detection_test,katz_test}/*/
     {input.cyp,test.yml} (harness float tolerance 1e-3,
     tests/mage/e2e/test_module.py:21). cugraph/online/subgraph cases are
     skipped (out of scope).
  2. The 12 exact PageRank vectors of
     /root/reference/src/mage/cpp/pagerank_module/pagerank_test.cpp:34-85
     (tolerances max-abs 1e-3 / avg-abs 1e-4, include/mg_test_utils.hpp:25-26),
     restated as data below and re-derived through oracle/_ref at gen time.

Every fixture is cross-checked at generation time: oracle restatement ==
reference-compiled core (_ref) == the published expected values.
"""
import json
import os
import re
import sys

import numpy as np
import yaml

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))
from oracle import Oracle, Reference  # noqa: E402

E2E = "/root/reference/tests/mage/e2e"
OUT_DIR = os.path.dirname(os.path.abspath(__file__))

NODE_RE = re.compile(r"\((\w+):Node\s*\{id:\s*(\d+)\}\)")
EDGE_RE = re.compile(r"\((\w+)\)-\[:(\w+)(?:\s*\{weight:\s*([0-9.eE+-]+)\})?\]->\((\w+)\)")

SUITES = {
    "pagerank_test": "pagerank",
    "weakly_connected_components_test": "wcc",
    "community_detection_test": "community_detection",
    "katz_test": "katz",
    "betweenness_centrality_test": "betweenness",
}
SKIP_SUBSTR = ("cugraph", "online", "subgraph")


def parse_cyp(path):
    """Parse the uniform MERGE/CREATE one-statement-per-line format into a
    dense graph: nodes in creation order (the skiplist scan order a live
    memgraphd would produce for sequential inserts)."""
    prop_to_dense = {}
    node_props = []
    src, dst, weights = [], [], []
    any_weight = False
    with open(path) as f:
        for line in f:
            alias = {}
            for m in NODE_RE.finditer(line):
                a, pid = m.group(1), int(m.group(2))
                alias[a] = pid
                if pid not in prop_to_dense:
                    prop_to_dense[pid] = len(node_props)
                    node_props.append(pid)
            for m in EDGE_RE.finditer(line):
                a_src, _, w, a_dst = m.group(1), m.group(2), m.group(3), m.group(4)
                src.append(prop_to_dense[alias[a_src]])
                dst.append(prop_to_dense[alias[a_dst]])
                weights.append(float(w) if w is not None else 1.0)
                if w is not None:
                    any_weight = True
    return node_props, src, dst, (weights if any_weight else None)


def parse_query(query):
    m = re.search(r"CALL\s+(\w+)\.(\w+)\(([^)]*)\)", query)
    mod, proc, raw_args = m.group(1), m.group(2), m.group(3).strip()
    args = []
    if raw_args:
        for tok in raw_args.split(","):
            tok = tok.strip()
            if tok in ("True", "true", "TRUE"):
                args.append(True)
            elif tok in ("False", "false", "FALSE"):
                args.append(False)
            elif tok.startswith('"') or tok.startswith("'"):
                args.append(tok.strip("\"'"))
            elif re.fullmatch(r"-?\d+", tok):
                args.append(int(tok))
            else:
                args.append(float(tok))
    rank_order = bool(re.search(r"ORDER BY\s+rank\s+DESC", query))
    return mod, proc, args, rank_order


def check_pagerank(o, ref, fx):
    args = fx["args"]
    max_iter = args[0] if len(args) > 0 else 100
    pr_o, _ = o.pagerank(fx["n_vertices"], fx["src"], fx["dst"], max_iterations=max_iter)
    pr_r = ref.pagerank(fx["n_vertices"], fx["src"], fx["dst"], max_iterations=max_iter)
    assert np.allclose(pr_o, pr_r, atol=1e-12), fx["name"]
    exp = {row["node"]: row["rank"] for row in fx["expected"]}
    for dense, pid in enumerate(fx["node_props"]):
        assert abs(pr_o[dense] - exp[pid]) < 1e-3, (fx["name"], pid, pr_o[dense], exp[pid])


def check_wcc(o, fx):
    comp, _ = o.wcc(fx["n_vertices"], fx["src"], fx["dst"])
    exp = {row["node_id"]: row["component_id"] for row in fx["expected"]}
    for dense, pid in enumerate(fx["node_props"]):
        assert comp[dense] == exp[pid], (fx["name"], pid, comp[dense], exp[pid])


def check_community(o, ref, fx):
    weight_prop_given = bool(fx["args"]) and isinstance(fx["args"][0], str)
    w = fx["weights"]
    c_o, _ = o.louvain(fx["n_vertices"], fx["src"], fx["dst"], weights=w)
    c_r = ref.louvain(fx["n_vertices"], fx["src"], fx["dst"], weights=w, n_threads=1)
    assert np.array_equal(c_o, c_r), fx["name"]
    exp = {row["node_id"]: row["community_id"] for row in fx["expected"]}
    for dense, pid in enumerate(fx["node_props"]):
        if pid in exp:
            assert c_o[dense] == exp[pid], (fx["name"], pid, c_o[dense], exp[pid])
    _ = weight_prop_given


def check_betweenness(o, ref, fx):
    args = fx["args"]
    directed = bool(args[0]) if len(args) > 0 else True
    normalize = bool(args[1]) if len(args) > 1 else True
    bc_o = o.betweenness(fx["n_vertices"], fx["src"], fx["dst"], directed, normalize)
    bc_r = ref.betweenness(fx["n_vertices"], fx["src"], fx["dst"], directed, normalize,
                           n_threads=1)
    assert np.abs(bc_o - bc_r).max() < 1e-9, fx["name"]
    exp = {row["node_id"]: row[[k for k in row if k != "node_id"][0]]
           for row in fx["expected"]}
    for dense, pid in enumerate(fx["node_props"]):
        if pid in exp:
            assert abs(bc_o[dense] - exp[pid]) < 1e-3, (fx["name"], pid, bc_o[dense])


def check_katz(o, ref, fx):
    alpha = fx["args"][0] if len(fx["args"]) > 0 else 0.2
    eps = fx["args"][1] if len(fx["args"]) > 1 else 1e-2
    k_o, _ = o.katz(fx["n_vertices"], fx["src"], fx["dst"], alpha=alpha, epsilon=eps)
    k_r = ref.katz(fx["n_vertices"], fx["src"], fx["dst"], alpha=alpha, epsilon=eps)
    assert np.allclose(k_o, k_r, atol=1e-12), fx["name"]
    # expected = node ids in (rank DESC, node.id ASC) order
    order = sorted(range(fx["n_vertices"]),
                   key=lambda d: (-k_o[d], fx["node_props"][d]))
    got = [fx["node_props"][d] for d in order]
    exp = [row["node_id"] for row in fx["expected"]]
    assert got == exp, (fx["name"], got, exp)


def gen_e2e():
    fixtures = []
    o, ref = Oracle(), Reference()
    for suite, algo in SUITES.items():
        for case in sorted(os.listdir(os.path.join(E2E, suite))):
            if any(s in case for s in SKIP_SUBSTR):
                continue
            d = os.path.join(E2E, suite, case)
            if not os.path.isdir(d):
                continue
            cyp = os.path.join(d, "input.cyp")
            yml = os.path.join(d, "test.yml")
            if not (os.path.exists(cyp) and os.path.exists(yml)):
                continue
            node_props, src, dst, weights = parse_cyp(cyp)
            with open(yml) as f:
                spec = yaml.safe_load(f)
            mod, proc, args, rank_order = parse_query(spec["query"])
            fx = {
                "name": f"{suite}/{case}",
                "algo": algo,
                "proc": proc,
                "args": args,
                "n_vertices": len(node_props),
                "node_props": node_props,
                "src": src,
                "dst": dst,
                "weights": weights,
                "expected": spec.get("output") or [],
                "rank_order": rank_order,
                "source": f"/root/reference/tests/mage/e2e/{suite}/{case}",
            }
            if fx["n_vertices"] > 0:
                if algo == "pagerank":
                    check_pagerank(o, ref, fx)
                elif algo == "wcc":
                    check_wcc(o, fx)
                elif algo == "community_detection":
                    check_community(o, ref, fx)
                elif algo == "katz":
                    check_katz(o, ref, fx)
                elif algo == "betweenness":
                    check_betweenness(o, ref, fx)
            fixtures.append(fx)
            print(f"ok {fx['name']}: V={fx['n_vertices']} E={len(src)}")
    with open(os.path.join(OUT_DIR, "e2e_cases.json"), "w") as f:
        json.dump(fixtures, f, indent=1)
    print(f"wrote {len(fixtures)} e2e fixtures")


# The 12 parametrized graphs + exact expected vectors of pagerank_test.cpp:34-85
# (defaults: max_iterations=100, damping=0.85, stop_epsilon=1e-5, 1 thread).
PAGERANK_UNIT = [
    (1, [], [1.00]),
    (2, [(0, 1)], [0.350877362, 0.649122638]),
    (0, [], []),
    (1, [(0, 0)], [1.00]),
    (2, [(0, 1), (0, 1)], [0.350877362, 0.649122638]),
    (2, [(1, 1)], [0.130435201, 0.869564799]),
    (5, [(0, 2), (0, 0), (2, 3), (3, 1), (1, 3), (1, 0), (1, 2), (3, 0), (0, 1), (3, 2)],
     [0.240963851, 0.187763717, 0.240963851, 0.294163985, 0.036144598]),
    (10, [(9, 5), (4, 4), (3, 8), (0, 5), (5, 0), (3, 0), (7, 9), (3, 9), (0, 4), (0, 4)],
     [0.114178360, 0.023587998, 0.023587998, 0.023587998, 0.588577186, 0.098712132,
      0.023587998, 0.023587998, 0.030271265, 0.050321066]),
    (10, [(8, 8), (2, 2), (0, 8), (7, 8), (1, 6), (0, 0), (1, 1), (6, 3), (9, 5)],
     [0.047683471, 0.047683471, 0.182781325, 0.067949168, 0.027417774, 0.050723042,
      0.047683471, 0.027417774, 0.473242731, 0.027417774]),
    (5, [(3, 3), (0, 3), (4, 2), (1, 1), (3, 2), (2, 0), (4, 0), (4, 4), (3, 2), (4, 1),
         (2, 4), (2, 2), (2, 3), (3, 3), (0, 0), (1, 0), (4, 2), (4, 0), (1, 2), (1, 4),
         (4, 0), (4, 0), (0, 0), (4, 0), (3, 3)],
     [0.304824023, 0.049593211, 0.217782046, 0.331928795, 0.095871925]),
    (4, [(1, 0), (3, 0), (2, 0), (3, 0)],
     [0.541985357, 0.152671548, 0.152671548, 0.152671548]),
    (7, [(0, 6), (3, 0), (6, 2), (0, 3), (2, 3), (6, 4), (1, 1), (2, 0), (0, 3), (5, 0),
         (0, 4), (5, 2), (1, 5), (5, 3), (2, 3), (6, 1), (2, 0), (6, 1), (2, 6), (2, 2),
         (0, 0), (6, 0), (6, 0), (0, 6), (3, 3), (6, 3), (1, 3), (4, 0), (1, 2), (2, 1)],
     [0.318471859, 0.075311781, 0.071307161, 0.295999683, 0.081155915, 0.037432346,
      0.120321254]),
]


def gen_pagerank_unit():
    o, ref = Oracle(), Reference()
    out = []
    for idx, (nv, edges, expected) in enumerate(PAGERANK_UNIT):
        src = [e[0] for e in edges]
        dst = [e[1] for e in edges]
        if nv > 0:
            pr_o, _ = o.pagerank(nv, src, dst)
            pr_r = ref.pagerank(nv, src, dst)
            assert np.allclose(pr_o, pr_r, atol=1e-12), idx
            exp = np.array(expected)
            assert np.abs(pr_o - exp).max() < 1e-3, (idx, pr_o, exp)
            assert np.abs(pr_o - exp).mean() < 1e-4, (idx, pr_o, exp)
        out.append({
            "name": f"pagerank_unit/case{idx:02d}",
            "n_vertices": nv,
            "src": src,
            "dst": dst,
            "expected_rank": expected,
            "source": "/root/reference/src/mage/cpp/pagerank_module/pagerank_test.cpp:34-85",
        })
        print(f"ok pagerank_unit/case{idx:02d}")
    with open(os.path.join(OUT_DIR, "pagerank_unit.json"), "w") as f:
        json.dump(out, f, indent=1)
    print(f"wrote {len(out)} pagerank unit fixtures")


if __name__ == "__main__":
    gen_e2e()
    gen_pagerank_unit()
