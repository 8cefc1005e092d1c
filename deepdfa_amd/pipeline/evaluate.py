"""Statement-level vulnerability labels (IVDetect method).

Parity target: reference sastvd/helpers/evaluate.py:19-255 — line-level
labels are the lines DATA- or CONTROL-dependent on the lines a fix ADDED
(get_dep_add_lines/get_dep_add_lines_bigvul), cached per dataset
(statement_labels.pkl). The reference extracts PDG edges with Joern; here
the dependency closure is computed from our CPG:

  * data dependence: reaching-definitions (analysis/dataflow.py) — a line
    using a variable depends on the lines whose definitions of it reach
    that statement;
  * control dependence (approximation without post-dominator trees): a
    statement with multiple CFG successors controls the statements
    reachable from it before the paths re-join.
"""

from __future__ import annotations

import pickle
from typing import Dict, List, Set

from ..analysis.dataflow import CPG, ReachingDefinitions


def data_dependences(cpg: CPG) -> Dict[int, Set[int]]:
    """node -> set of defining nodes whose definitions it uses."""
    rd = ReachingDefinitions(cpg)
    IN, _OUT = rd.solve()
    ast = cpg.edge_subgraph("AST")
    deps: Dict[int, Set[int]] = {}
    for nid in cpg.nodes:
        used: Set[str] = set()
        stack = list(ast.get(nid, []))
        while stack:
            c = stack.pop()
            if cpg.nodes[c].get("_label") == "IDENTIFIER":
                used.add(cpg.nodes[c].get("name"))
            stack.extend(ast.get(c, []))
        deps[nid] = {d.node for d in IN.get(nid, set()) if d.v in used}
    return deps


def control_dependences(cpg: CPG) -> Dict[int, Set[int]]:
    """node -> branch nodes it is (approximately) control-dependent on."""
    succ = cpg.edge_subgraph("CFG")
    branches = [n for n, ss in succ.items() if len(ss) > 1]
    deps: Dict[int, Set[int]] = {n: set() for n in cpg.nodes}
    for b in branches:
        # nodes reachable from b within a bounded window depend on b
        seen, frontier = set(), list(succ[b])
        for _ in range(16):
            nxt = []
            for n in frontier:
                if n in seen:
                    continue
                seen.add(n)
                deps[n].add(b)
                nxt.extend(succ.get(n, []))
            frontier = nxt
    return deps


def feature_extraction(cpg: CPG):
    """Returns (line_of_node, combined dependency map node -> nodes)."""
    dd = data_dependences(cpg)
    cd = control_dependences(cpg)
    lines = {n: cpg.nodes[n].get("lineNumber", -1) for n in cpg.nodes}
    combined = {n: dd.get(n, set()) | cd.get(n, set()) for n in cpg.nodes}
    return lines, combined


def get_dep_add_lines(cpg: CPG, added_lines: Set[int]) -> Set[int]:
    """Lines dependent on any ADDED line (evaluate.py:194-218): a statement
    is vulnerable-relevant if it depends on a node whose line was added, or
    its own line was added."""
    lines, deps = feature_extraction(cpg)
    added_nodes = {n for n, ln in lines.items() if ln in added_lines}
    out = set(added_lines)
    for n, dset in deps.items():
        if dset & added_nodes:
            out.add(lines[n])
    return {ln for ln in out if ln >= 0}


def get_dep_add_lines_bigvul(
    cpgs: Dict[int, CPG], added: Dict[int, Set[int]], cache_path: str = None
) -> Dict[int, List[int]]:
    """Per-example dependent-line labels with pickle caching
    (statement_labels.pkl contract)."""
    out = {}
    for _id, cpg in cpgs.items():
        out[_id] = sorted(get_dep_add_lines(cpg, added.get(_id, set())))
    if cache_path:
        with open(cache_path, "wb") as f:
            pickle.dump(out, f)
    return out
