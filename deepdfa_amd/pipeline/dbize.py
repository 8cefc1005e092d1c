"""Databaseization: CPGs -> tabular artifacts -> batched-CFG graph files.

Parity target: reference sastvd/scripts/dbize.py:30-105 (nodes.csv /
edges.csv with per-node vuln labels), dbize_graphs.py:20-33 (edge CSV ->
graph + self-loops -> graphs.bin with graph ids) and dbize_absdf.py
(per-subkey feature index CSVs for each limit_all). Output layout:

  <out_dir>/nodes.csv                       (graph_id, node_idx, lineNumber, vuln)
  <out_dir>/edges.csv                       (graph_id, src, dst, etype)
  <out_dir>/graphs/<id>.pt                  (BatchedCFG with ndata)
  <out_dir>/nodes_feat_<FEATNAME>.csv       (graph_id, node_idx, feat_idx)

The graphs/<id>.pt files are directly consumable by
BigVulDatasetLineVD(graph_dir=...).
"""

from __future__ import annotations

import os
from typing import Dict, Iterable, List, Optional, Set

import pandas as pd
import torch

from ..analysis.dataflow import CPG
from ..data.features import parse_limits
from ..graph import BatchedCFG
from .absdf import SUBKEYS, build_vocab, get_dataflow_features, node_feature_indices


def cfg_nodes(cpg: CPG) -> List[int]:
    """CFG-participating statement nodes in deterministic order."""
    in_cfg: Set[int] = set()
    for s, d, t in cpg.edges:
        if t == "CFG":
            in_cfg.add(s)
            in_cfg.add(d)
    return sorted(in_cfg)


def cpg_to_tables(cpg: CPG, graph_id: int, vuln_lines: Optional[Set[int]] = None):
    """nodes/edges dataframes over the CFG statement nodes; node vuln label
    = its line is in vuln_lines (dbize.py get_vuln :35-39 semantics)."""
    order = cfg_nodes(cpg)
    pos = {nid: i for i, nid in enumerate(order)}
    vuln_lines = vuln_lines or set()
    nodes = pd.DataFrame(
        {
            "graph_id": graph_id,
            "node_idx": range(len(order)),
            "cpg_id": order,
            "lineNumber": [cpg.nodes[n].get("lineNumber", -1) for n in order],
            "code": [cpg.nodes[n].get("code", "") for n in order],
            "vuln": [int(cpg.nodes[n].get("lineNumber", -1) in vuln_lines) for n in order],
        }
    )
    e_rows = [
        {"graph_id": graph_id, "src": pos[s], "dst": pos[d], "etype": t}
        for s, d, t in cpg.edges
        if t == "CFG" and s in pos and d in pos
    ]
    edges = pd.DataFrame(e_rows, columns=["graph_id", "src", "dst", "etype"])
    return nodes, edges, order


def build_graph(
    nodes: pd.DataFrame, edges: pd.DataFrame, feats: Dict[str, List[int]]
) -> BatchedCFG:
    """One BatchedCFG with self-loops (dbize_graphs.py:25) and ndata."""
    n = len(nodes)
    ndata = {"_VULN": torch.tensor(nodes.vuln.values, dtype=torch.int64)}
    for name, idx in feats.items():
        ndata[name] = torch.tensor(idx, dtype=torch.int64)
    return BatchedCFG.from_edges(
        n, edges.src.tolist(), edges.dst.tolist(), ndata=ndata, add_self_loops=True
    )


def dbize(
    cpgs: Dict[int, CPG],
    out_dir: str,
    train_ids: Iterable[int],
    feat: str = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000",
    vuln_lines: Optional[Dict[int, Set[int]]] = None,
):
    """Full stages 3-5 of the reference preprocess.sh: tables, features,
    vocab from the train split, graph files. Returns the vocab."""
    spec = parse_limits(feat)
    os.makedirs(os.path.join(out_dir, "graphs"), exist_ok=True)
    vuln_lines = vuln_lines or {}
    all_nodes, all_edges = [], []
    features = {gid: get_dataflow_features(cpg) for gid, cpg in cpgs.items()}
    train_feats = [features[g] for g in train_ids if g in features and len(features[g])]
    train_df = (
        pd.concat(train_feats, ignore_index=True)
        if train_feats
        else pd.DataFrame(columns=["node_id"] + SUBKEYS)
    )
    vocabs = {sk: build_vocab(train_df, [sk], spec.limit_all) for sk in SUBKEYS}
    feat_rows = {sk: [] for sk in SUBKEYS}
    for gid, cpg in cpgs.items():
        nodes, edges, order = cpg_to_tables(cpg, gid, vuln_lines.get(gid))
        all_nodes.append(nodes)
        all_edges.append(edges)
        feats = {}
        for sk in SUBKEYS:
            idx = node_feature_indices(
                features[gid], vocabs[sk], [sk], len(order), node_order=order
            )
            feats[f"_ABS_DATAFLOW_{sk}"] = idx
            feat_rows[sk] += [
                {"graph_id": gid, "node_idx": i, "feat_idx": v} for i, v in enumerate(idx)
            ]
        g = build_graph(nodes, edges, feats)
        g.save(os.path.join(out_dir, "graphs", f"{gid}.pt"))
    pd.concat(all_nodes, ignore_index=True).to_csv(os.path.join(out_dir, "nodes.csv"), index=False)
    pd.concat(all_edges, ignore_index=True).to_csv(os.path.join(out_dir, "edges.csv"), index=False)
    for sk in SUBKEYS:
        pd.DataFrame(feat_rows[sk]).to_csv(
            os.path.join(
                out_dir,
                f"nodes_feat__ABS_DATAFLOW_{sk}_all_limitall_{spec.limit_all}"
                f"_limitsubkeys_{spec.limit_subkeys}_fixed.csv",
            ),
            index=False,
        )
    return vocabs


def load_graphs_from_csv(out_dir: str, feat_names=None):
    """Graph loader over the CSV artifacts (reference
    sastvd/linevd/graphmogrifier.py:20-97 semantics: merge nodes.csv with
    the per-feature nodes_feat_*.csv tables, rebuild per-graph BatchedCFGs
    from edges.csv with self-loops, attach _ABS_DATAFLOW_* and _VULN).
    The graphs/<id>.pt fast path bakes the same ndata at dbize time; this
    loader exists for artifact dirs that only carry the CSV tables."""
    import glob as _glob

    nodes = pd.read_csv(os.path.join(out_dir, "nodes.csv"))
    edges = pd.read_csv(os.path.join(out_dir, "edges.csv"))
    feats = {}
    for path in _glob.glob(os.path.join(out_dir, "nodes_feat_*.csv")):
        base = os.path.basename(path)[len("nodes_feat_"):-len(".csv")]
        # name like _ABS_DATAFLOW_api_all_limitall_1000_..._fixed
        featname = "_".join(base.split("_")[:4])  # _ABS_DATAFLOW_<subkey>
        feats[featname] = pd.read_csv(path)
    graphs = {}
    for gid, nd in nodes.groupby("graph_id"):
        nd = nd.sort_values("node_idx")
        n = len(nd)
        ed = edges[edges.graph_id == gid]
        ndata = {"_VULN": torch.tensor(nd.vuln.values, dtype=torch.int64)}
        for featname, fdf in feats.items():
            sub = fdf[fdf.graph_id == gid].sort_values("node_idx")
            ndata[featname] = torch.tensor(sub.feat_idx.values, dtype=torch.int64)
        graphs[int(gid)] = BatchedCFG.from_edges(
            n, ed.src.tolist(), ed.dst.tolist(), ndata=ndata, add_self_loops=True
        )
    return graphs
