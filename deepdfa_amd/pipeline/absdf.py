"""Abstract-dataflow feature extraction + vocabulary build.

Parity target: reference DDFA/sastvd/scripts/abstract_dataflow_full.py
(get_dataflow_features :54-200, to_hash :285-295, stage-2 hash grouping)
and sastvd/helpers/datasets.py:587-692 (abs_dataflow: top-limit_all hashes
from the TRAIN split become the index; 0 = "not a definition", 1 = UNKNOWN,
matching dbize_absdf.py:35-42) plus dbize_absdf.py's per-subkey index CSV
export.

Per assignment/inc-dec statement, four subkey sets are extracted from its
AST:
  api       — names of non-operator CALLs in the statement
  datatype  — declared type of the defined variable
  literal   — literal tokens in the statement
  operator  — operator CALL names in the statement
"""

from __future__ import annotations

from collections import Counter
from typing import Dict, List, Optional

import pandas as pd

from .. import hashstr
from ..analysis.dataflow import CPG, MOD_OPS, defined_variable

SUBKEYS = ["api", "datatype", "literal", "operator"]


def get_dataflow_features(cpg: CPG) -> pd.DataFrame:
    """Rows: (node_id, api, datatype, literal, operator) for every
    definition statement in the CPG; subkey values are sorted
    '|'-joined strings (sets)."""
    ast = cpg.edge_subgraph("AST")
    rows = []
    for nid, n in cpg.nodes.items():
        if n.get("_label") != "CALL" or n.get("name") not in MOD_OPS:
            continue
        v = defined_variable(cpg, nid)
        api, lit, oper = set(), set(), set()
        datatype = set()
        stack = list(ast.get(nid, []))
        while stack:
            c = stack.pop()
            cn = cpg.nodes[c]
            lab = cn.get("_label")
            if lab == "CALL":
                name = cn.get("name", "")
                if name.startswith("<operator"):
                    if name != "<operator>.assignment":
                        oper.add(name)
                else:
                    api.add(name)
            elif lab == "LITERAL":
                lit.add(str(cn.get("code", cn.get("name", ""))))
            elif lab == "IDENTIFIER" and cn.get("name") == v and cn.get("typeFullName"):
                datatype.add(cn["typeFullName"])
            stack.extend(ast.get(c, []))
        rows.append(
            {
                "node_id": nid,
                "api": "|".join(sorted(api)),
                "datatype": "|".join(sorted(datatype)),
                "literal": "|".join(sorted(lit)),
                "operator": "|".join(sorted(oper)),
            }
        )
    return pd.DataFrame(rows, columns=["node_id"] + SUBKEYS)


def to_hash(row, subkeys: List[str]) -> str:
    """Stable statement hash over the selected subkeys
    (abstract_dataflow_full.py:285-295 contract)."""
    parts = [f"{k}:{row[k]}" for k in subkeys]
    return str(hashstr("//".join(parts)))


def build_vocab(
    train_features: pd.DataFrame, subkeys: List[str], limit_all: int
) -> Dict[str, int]:
    """Top-limit_all hashes from the TRAIN split -> index starting at 2
    (0 = not-a-definition, 1 = UNKNOWN)."""
    hashes = [to_hash(r, subkeys) for _, r in train_features.iterrows()]
    counts = Counter(hashes)
    vocab = {}
    for i, (h, _c) in enumerate(counts.most_common(limit_all)):
        vocab[h] = i + 2
    return vocab


def node_feature_indices(
    features: pd.DataFrame, vocab: Dict[str, int], subkeys: List[str], num_nodes: int,
    node_order: Optional[List[int]] = None,
) -> List[int]:
    """Per-node feature index list: 0 for non-definition nodes, 1 for
    definitions whose hash is out-of-vocabulary (dbize_absdf.py:35-42)."""
    idx_by_node = {}
    for _, r in features.iterrows():
        h = to_hash(r, subkeys)
        idx_by_node[int(r.node_id)] = vocab.get(h, 1)
    if node_order is None:
        node_order = list(range(num_nodes))
    return [idx_by_node.get(n, 0) for n in node_order]
