"""CPG construction: Joern JSON parsing + synthetic generator.

Parity targets:
  * reference sastvd/helpers/joern.py:182-319 (get_node_edges): parse
    `<file>.nodes.json` / `<file>.edges.json` exported by the Joern script
    into node/edge tables, dropping COMMENT/FILE nodes and
    CONTAINS/DOMINATE/POST_DOMINATE/SOURCE_FILE edges (:251-258);
  * a synthetic CPG generator that emits the same structure (CFG/AST/
    ARGUMENT edges over METHOD/CALL/IDENTIFIER/LITERAL/BLOCK nodes) from
    the deterministic pseudo-C functions — the no-Joern test/bench path
    with ground-truth dataflow.
"""

from __future__ import annotations

import json
from typing import Dict, List, Tuple

import numpy as np

from ..analysis.dataflow import CPG

DROP_NODE_LABELS = {"COMMENT", "FILE"}
DROP_EDGE_TYPES = {"CONTAINS", "DOMINATE", "POST_DOMINATE", "SOURCE_FILE"}


def parse_joern_json(nodes_json: str, edges_json: str) -> CPG:
    """Parse Joern export files (get_func_graph.sc output format)."""
    with open(nodes_json) as f:
        raw_nodes = json.load(f)
    with open(edges_json) as f:
        raw_edges = json.load(f)
    nodes: Dict[int, Dict] = {}
    for n in raw_nodes:
        if isinstance(n, list):  # [id, {props}] variant
            nid, props = n[0], dict(n[1])
        else:
            props = dict(n)
            nid = props.get("id")
        if props.get("_label") in DROP_NODE_LABELS:
            continue
        nodes[int(nid)] = props
    edges: List[Tuple[int, int, str]] = []
    for e in raw_edges:
        if isinstance(e, dict):
            s, d, t = e.get("inNode") or e.get("src"), e.get("outNode") or e.get("dst"), e.get("etype") or e.get("_label")
        else:
            s, d, t = e[0], e[1], e[2]
        if t in DROP_EDGE_TYPES:
            continue
        if int(s) in nodes and int(d) in nodes:
            edges.append((int(s), int(d), str(t)))
    return CPG(nodes, edges)


def synthetic_cpg(_id: int, n_stmts: int = 12) -> CPG:
    """Deterministic CPG for pseudo-C function `_id`: a chain of statements
    (assignments, calls, inc/dec, branches) with full CFG/AST/ARGUMENT
    structure. Statement list mirrors data/text_dataset.synthetic_func_source
    in spirit; here the AST is built directly so the abstract-dataflow
    features and reaching-definitions are well-defined."""
    rng = np.random.RandomState(_id % (2**31))
    names = ["buf", "len", "ptr", "data", "size", "idx", "tmp", "out", "src", "dst"]
    types = ["int", "char*", "size_t", "uint32_t", "void*"]
    apis = ["memcpy", "strlen", "malloc", "read_input", "check_bounds"]
    nodes: Dict[int, Dict] = {}
    edges: List[Tuple[int, int, str]] = []
    nid = [0]

    def add(label, **props):
        nodes[nid[0]] = {"_label": label, **props}
        nid[0] += 1
        return nid[0] - 1

    method = add("METHOD", name=f"func_{_id}", code=f"func_{_id}", lineNumber=1)
    var_types = {n: types[rng.randint(len(types))] for n in names}
    stmt_nodes = []
    for line in range(2, 2 + n_stmts):
        a, b, c = (names[rng.randint(10)] for _ in range(3))
        r = rng.rand()
        if r < 0.45:  # a = b <op> c
            op = ["<operator>.addition", "<operator>.subtraction",
                  "<operator>.multiplication"][rng.randint(3)]
            call = add("CALL", name="<operator>.assignment",
                       code=f"{a} = {b} op {c}", lineNumber=line)
            lhs = add("IDENTIFIER", name=a, code=a, order=1,
                      typeFullName=var_types[a], lineNumber=line)
            rhs = add("CALL", name=op, code=f"{b} op {c}", order=2, lineNumber=line)
            b1 = add("IDENTIFIER", name=b, code=b, order=1, lineNumber=line)
            b2 = add("IDENTIFIER", name=c, code=c, order=2, lineNumber=line)
            edges += [(call, lhs, "AST"), (call, rhs, "AST"), (rhs, b1, "AST"),
                      (rhs, b2, "AST"), (call, lhs, "ARGUMENT"), (call, rhs, "ARGUMENT")]
        elif r < 0.6:  # a = api(b, LITERAL)
            api = apis[rng.randint(len(apis))]
            call = add("CALL", name="<operator>.assignment",
                       code=f"{a} = {api}(...)", lineNumber=line)
            lhs = add("IDENTIFIER", name=a, code=a, order=1,
                      typeFullName=var_types[a], lineNumber=line)
            rhs = add("CALL", name=api, code=f"{api}(...)", order=2, lineNumber=line)
            arg1 = add("IDENTIFIER", name=b, code=b, order=1, lineNumber=line)
            lit = add("LITERAL", name=str(rng.randint(256)), code=str(rng.randint(256)),
                      order=2, lineNumber=line)
            edges += [(call, lhs, "AST"), (call, rhs, "AST"), (rhs, arg1, "AST"),
                      (rhs, lit, "AST"), (call, lhs, "ARGUMENT"), (call, rhs, "ARGUMENT")]
        elif r < 0.75:  # a++
            call = add("CALL", name="<operator>.postIncrement", code=f"{a}++", lineNumber=line)
            lhs = add("IDENTIFIER", name=a, code=a, order=1,
                      typeFullName=var_types[a], lineNumber=line)
            edges += [(call, lhs, "AST"), (call, lhs, "ARGUMENT")]
        else:  # condition (no definition)
            call = add("CALL", name="<operator>.lessThan", code=f"{a} < {b}", lineNumber=line)
            b1 = add("IDENTIFIER", name=a, code=a, order=1, lineNumber=line)
            b2 = add("IDENTIFIER", name=b, code=b, order=2, lineNumber=line)
            edges += [(call, b1, "AST"), (call, b2, "AST")]
        stmt_nodes.append(call)

    # CFG: mostly sequential with occasional branch joins/back edges
    edges.append((method, stmt_nodes[0], "CFG"))
    for i in range(1, len(stmt_nodes)):
        edges.append((stmt_nodes[i - 1], stmt_nodes[i], "CFG"))
        rr = rng.rand()
        if rr < 0.2 and i + 2 < len(stmt_nodes):
            edges.append((stmt_nodes[i - 1], stmt_nodes[min(i + 2, len(stmt_nodes) - 1)], "CFG"))
        elif rr < 0.28 and i > 2:
            edges.append((stmt_nodes[i], stmt_nodes[rng.randint(0, i - 1)], "CFG"))
    for s in stmt_nodes:
        edges.append((method, s, "AST"))
    return CPG(nodes, edges)


# graph reduction by edge-type set (reference sastvd/helpers/joern.py:419-441
# rdg): which edge types each gtype keeps
RDG_TYPES = {
    "reftype": {"EVAL_TYPE", "REF"},
    "ast": {"AST"},
    "pdg": {"REACHING_DEF", "CDG"},
    "cfgcdg": {"CFG", "CDG"},
    "cfg": {"CFG"},
    "all": {"REACHING_DEF", "CDG", "AST", "EVAL_TYPE", "REF"},
    "dataflow": {"CFG", "AST"},
}


def rdg(cpg: CPG, gtype: str) -> CPG:
    """Reduce the CPG to the edge types of `gtype` (joern.py rdg parity)."""
    keep = RDG_TYPES[gtype]
    return CPG(dict(cpg.nodes), [e for e in cpg.edges if e[2] in keep])


def drop_lone_nodes(cpg: CPG) -> CPG:
    """Remove nodes with no edge connections (joern.py:485-493)."""
    touched = set()
    for s, d, _ in cpg.edges:
        touched.add(s)
        touched.add(d)
    return CPG({i: p for i, p in cpg.nodes.items() if i in touched}, list(cpg.edges))


def group_nodes_by_line(cpg: CPG) -> CPG:
    """Statement-level grouping (reference sastvd/linevd/utils.py:6-22
    ne_groupnodes): collapse all CPG nodes sharing a lineNumber into one
    statement node (the first), re-target edges to line numbers, dedupe,
    and drop nodes with no line. Node ids in the result ARE line numbers.
    The real-Joern path runs this before CFG extraction; the synthetic
    generator already emits one node per statement."""
    line_of = {}
    rep: Dict[int, Dict] = {}
    for nid in sorted(cpg.nodes):
        props = cpg.nodes[nid]
        ln = props.get("lineNumber", -1)
        if ln is None or ln < 0:
            continue
        line_of[nid] = ln
        if ln not in rep:
            rep[ln] = dict(props)
    edges = []
    seen = set()
    for s, d, t in cpg.edges:
        if s not in line_of or d not in line_of:
            continue
        e = (line_of[s], line_of[d], t)
        if e not in seen and e[0] != e[1]:
            seen.add(e)
            edges.append(e)
    touched = {x for s, d, _ in edges for x in (s, d)}
    return CPG({ln: p for ln, p in rep.items() if ln in touched}, edges)
