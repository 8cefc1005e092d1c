"""Joern CPG graph helpers over node/edge tables.

Parity targets (reference sastvd/helpers/joern.py):
  * rdg (:419-441): reduce an edge table to a graph type's edge subset —
    same gtype vocabulary (reftype/ast/pdg/cfgcdg/cfg/all/dataflow);
  * neighbour_nodes (:372-416): k-hop neighbourhoods via sparse adjacency
    powers (scipy CSR), undirected, with/without intermediate hops;
  * assign_line_num_to_local (:444-482): LOCAL variable declarations carry
    no line number in Joern's export; recover it from the 1-hop AST block,
    the 2-hop reftype TYPE node, and a text search for "<type><name>;"
    below the block's line;
  * export of the ReachingDef dataflow solution (get_dataflow_output.sc:
    25-72 capability): the reference runs Joern's own dataflow engine and
    dumps gen/kill/in/out per node — here the native Kildall solver
    (analysis/dataflow.py) produces the same JSON shape, so downstream
    consumers work with or without the JVM.

Tables: nodes(id, _label, name, lineNumber, ...), edges(innode, outnode,
etype) — the get_node_edges export schema.
"""

from __future__ import annotations

import json
from collections import defaultdict
from typing import Dict, List, Optional

import numpy as np
import pandas as pd
from scipy import sparse


def rdg(edges: pd.DataFrame, gtype: str) -> pd.DataFrame:
    """Reduce the edge table to one graph type (reference joern.py:419-441)."""
    et = edges.etype
    if gtype == "reftype":
        return edges[(et == "EVAL_TYPE") | (et == "REF")]
    if gtype == "ast":
        return edges[et == "AST"]
    if gtype == "pdg":
        return edges[(et == "REACHING_DEF") | (et == "CDG")]
    if gtype == "cfgcdg":
        return edges[(et == "CFG") | (et == "CDG")]
    if gtype == "cfg":
        return edges[et == "CFG"]
    if gtype == "all":
        return edges[
            (et == "REACHING_DEF") | (et == "CDG") | (et == "AST")
            | (et == "EVAL_TYPE") | (et == "REF")
        ]
    if gtype == "dataflow":
        return edges[(et == "CFG") | (et == "AST")]
    raise ValueError(f"unknown gtype {gtype!r}")


def neighbour_nodes(
    nodes: pd.DataFrame,
    edges: pd.DataFrame,
    nodeids: List[int],
    hop: int = 1,
    intermediate: bool = True,
) -> Dict[int, List[int]]:
    """k-hop neighbour ids per query node over the UNDIRECTED edge set,
    via sparse adjacency matrix powers (reference joern.py:372-416)."""
    ids = nodes.id.to_numpy()
    id2adj = {int(i): a for a, i in enumerate(ids)}
    src = edges.innode.map(id2adj).to_numpy()
    dst = edges.outnode.map(id2adj).to_numpy()
    n = len(ids)
    neighbours: Dict[int, List[int]] = defaultdict(list)
    if len(src) == 0:
        for nid in nodeids:
            neighbours[int(nid)] = []
        return neighbours
    rows = np.concatenate([src, dst])
    cols = np.concatenate([dst, src])
    coo = sparse.coo_matrix((np.ones(len(rows)), (rows, cols)), shape=(n, n))
    base = coo.tocsr()

    def collect(csr, nid):
        a = id2adj.get(int(nid))
        if a is None:
            return []
        return [int(ids[j]) for j in csr.getrow(a).nonzero()[1]]

    if intermediate:
        acc = base.copy()
        for h in range(1, hop + 1):
            for nid in nodeids:
                neighbours[int(nid)] += collect(acc, nid)
            if h < hop:
                acc = acc @ base
    else:
        acc = base
        for _ in range(hop - 1):
            acc = acc @ base
        for nid in nodeids:
            neighbours[int(nid)] += collect(acc, nid)
    return neighbours


def assign_line_num_to_local(
    nodes: pd.DataFrame, edges: pd.DataFrame, code: List[str],
) -> Dict[int, int]:
    """Recover line numbers for LOCAL declaration nodes
    (reference joern.py:444-482 algorithm): the TYPE reached in 2 reftype
    hops names the declared type; the 1-hop AST parent BLOCK bounds the
    search; the first source line below it whose whitespace-stripped text
    equals "<type><name>;" is the declaration line (1-based)."""
    local_ids = nodes[nodes._label == "LOCAL"].id.tolist()
    if not local_ids:
        return {}
    onehop = neighbour_nodes(nodes, rdg(edges, "ast"), local_ids, 1, False)
    twohop = neighbour_nodes(nodes, rdg(edges, "reftype"), local_ids, 2, False)
    types = nodes[nodes._label == "TYPE"]
    id2name = dict(zip(types.id, types.name))
    blocks = nodes[(nodes._label == "BLOCK") | (nodes._label == "CONTROL_STRUCTURE")]
    block2line = dict(zip(blocks.id, blocks.lineNumber))
    id2local = dict(zip(nodes.id, nodes.name))
    stripped = ["".join(str(line).split()) for line in code]
    out: Dict[int, int] = {}
    for nid in local_ids:
        type_ids = [i for i in twohop.get(nid, []) if i in id2name]
        block_ids = [i for i in onehop.get(nid, []) if i in block2line]
        if not type_ids or not block_ids:
            continue
        tname = id2name[type_ids[0]]
        bline = block2line[block_ids[0]]
        if bline is None or (isinstance(bline, float) and np.isnan(bline)):
            continue
        target = "".join((str(tname) + str(id2local.get(nid, ""))).split()) + ";"
        start = int(bline)
        try:
            rel = stripped[start:].index(target)
        except ValueError:
            continue
        out[int(nid)] = start + rel + 1
    return out


# -- ReachingDef solution export (get_dataflow_output.sc capability) --------

def export_dataflow_solution(cpg, path: Optional[str] = None) -> Dict:
    """Solve reaching definitions on the CPG with the native Kildall
    worklist (analysis/dataflow.py) and emit the reference's
    <file>.dataflow.json shape: per CFG node, the gen/kill sets and the
    in/out solution (definitions as [variable, def-node] pairs)."""
    from ..analysis.dataflow import ReachingDefinitions

    rd = ReachingDefinitions(cpg)
    sol_in, sol_out = rd.solve()

    def pairs(defs):
        return sorted([d.v, str(d.node)] for d in defs)

    out = {
        "nodes": {
            str(n): {
                "gen": pairs(rd.gen[n]),
                "kill": pairs(rd.kill(n)),
                "in": pairs(sol_in.get(n, set())),
                "out": pairs(sol_out.get(n, set())),
            }
            for n in cpg.nodes
        }
    }
    if path is not None:
        with open(path, "w") as f:
            json.dump(out, f)
    return out
