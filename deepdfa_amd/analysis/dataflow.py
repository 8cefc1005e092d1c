"""Reaching-definitions dataflow analysis (Kildall worklist).

Parity target: reference DDFA/code_gnn/analysis/dataflow.py:103-250 — the
pure-Python reaching-definitions solver over a function's CPG that DeepDFA's
GGNN is trained to imitate. It is the framework's semantic oracle: the
worklist algorithm computes, per CFG node, the set of variable definitions
reaching it (gen/kill over assignment and inc/dec operators).

This implementation works over a lightweight CPG structure (node table with
Joern-style `_label`/`name`/`code` fields + CFG/AST/ARGUMENT edges), which
the preprocessing pipeline produces from Joern JSON exports or from the
synthetic C generator.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Set, Tuple

ASSIGNMENT_OPS = [
    "<operator>.assignment",
    "<operator>.assignmentAnd",
    "<operator>.assignmentArithmeticShiftRight",
    "<operator>.assignmentDivision",
    "<operator>.assignmentExponentiation",
    "<operator>.assignmentLogicalShiftRight",
    "<operator>.assignmentMinus",
    "<operator>.assignmentModulo",
    "<operator>.assignmentMultiplication",
    "<operator>.assignmentOr",
    "<operator>.assignmentPlus",
    "<operator>.assignmentShiftLeft",
    "<operator>.assignmentXor",
]
INC_DEC_OPS = [
    "<operator>.incBy",
    "<operator>.postDecrement",
    "<operator>.postIncrement",
    "<operator>.preDecrement",
    "<operator>.preIncrement",
]
MOD_OPS = ASSIGNMENT_OPS + INC_DEC_OPS
MOD_OPS += [op.replace("<operator>", "<operators>") for op in ASSIGNMENT_OPS + INC_DEC_OPS]


@dataclass(frozen=True)
class VariableDefinition:
    """One definition site: variable name + defining CFG node."""

    v: str
    node: int
    code: str = field(default="", compare=False, hash=False)


@dataclass
class CPG:
    """Minimal code-property-graph: typed nodes + typed directed edges."""

    nodes: Dict[int, Dict]  # id -> {"_label", "name", "code", "lineNumber", ...}
    edges: List[Tuple[int, int, str]]  # (src, dst, etype in {CFG, AST, ARGUMENT, ...})

    def edge_subgraph(self, etype: str) -> Dict[int, List[int]]:
        adj: Dict[int, List[int]] = {n: [] for n in self.nodes}
        for s, d, t in self.edges:
            if t == etype:
                adj[s].append(d)
        return adj

    def predecessors(self, etype: str) -> Dict[int, List[int]]:
        pred: Dict[int, List[int]] = {n: [] for n in self.nodes}
        for s, d, t in self.edges:
            if t == etype:
                pred[d].append(s)
        return pred


def defined_variable(cpg: CPG, node_id: int) -> Optional[str]:
    """If `node_id` is a call to a modifying operator, the name of the
    variable it (re)defines: the identifier under its first AST argument
    (reference dataflow.py get_gen semantics)."""
    n = cpg.nodes[node_id]
    if n.get("_label") != "CALL" or n.get("name") not in MOD_OPS:
        return None
    ast = cpg.edge_subgraph("AST")
    arg_order = {
        c: cpg.nodes[c].get("order", i) for i, c in enumerate(ast.get(node_id, []))
    }
    children = sorted(ast.get(node_id, []), key=lambda c: arg_order[c])
    if not children:
        return None
    first = children[0]
    # descend to an identifier
    stack = [first]
    while stack:
        c = stack.pop(0)
        if cpg.nodes[c].get("_label") == "IDENTIFIER":
            return cpg.nodes[c].get("name")
        stack = ast.get(c, []) + stack
    return None


class ReachingDefinitions:
    """Kildall worklist solver (reference dataflow.py:103-177)."""

    def __init__(self, cpg: CPG):
        self.cpg = cpg
        self.cfg_succ = cpg.edge_subgraph("CFG")
        self.cfg_pred = cpg.predecessors("CFG")
        # domain: every definition site
        self.domain: List[VariableDefinition] = []
        self.gen: Dict[int, Set[VariableDefinition]] = {}
        for nid in cpg.nodes:
            v = defined_variable(cpg, nid)
            if v is not None:
                d = VariableDefinition(v, nid, cpg.nodes[nid].get("code", ""))
                self.domain.append(d)
                self.gen[nid] = {d}
            else:
                self.gen[nid] = set()

    def kill(self, nid: int) -> Set[VariableDefinition]:
        g = self.gen[nid]
        if not g:
            return set()
        names = {d.v for d in g}
        return {d for d in self.domain if d.v in names and d.node != nid}

    def solve(self) -> Tuple[Dict[int, Set], Dict[int, Set]]:
        """Returns (IN, OUT) per node."""
        IN: Dict[int, Set] = {n: set() for n in self.cpg.nodes}
        OUT: Dict[int, Set] = {n: set() for n in self.cpg.nodes}
        work = list(self.cpg.nodes)
        while work:
            n = work.pop(0)
            new_in: Set = set()
            for p in self.cfg_pred[n]:
                new_in |= OUT[p]
            new_out = self.gen[n] | (new_in - self.kill(n))
            if new_in != IN[n] or new_out != OUT[n]:
                IN[n], OUT[n] = new_in, new_out
                for s in self.cfg_succ[n]:
                    if s not in work:
                        work.append(s)
        return IN, OUT

    def get_reaching_definitions(self):
        return self.solve()
