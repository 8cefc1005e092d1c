"""Graph dataset: per-partition CFGs with abstract-dataflow features.

Parity target: reference sastvd/linevd/dataset.py:13-76
(BigVulDatasetLineVD: `item` returns (graph, extrafeats); `get_indices`
batches graphs by example id for the combined models, silently dropping
examples whose CFG failed preprocessing) and sastvd/linevd/graphmogrifier.py
(attaching _ABS_DATAFLOW_* / _VULN ndata).

Graph sources:
  * real artifacts: a directory of saved BatchedCFG files + nodes feature
    CSVs (the dbize pipeline's output format, deepdfa_amd/data/pipeline.py);
  * synthetic: deterministic per-id generation (seeded by example id), the
    no-network benchmarking path required by BASELINE.json.
"""

from __future__ import annotations

import os
from typing import Dict, List, Optional, Sequence, Tuple

import torch

from ..graph import BatchedCFG, batch_graphs
from ..graph.synthetic import synthetic_cfg
from .dclass import BigVulDataset
from .features import parse_limits


class BigVulDatasetLineVD(BigVulDataset):
    """Returns (BatchedCFG, extrafeats) per example."""

    def __init__(
        self,
        feat: str = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000",
        gtype: str = "cfg",
        graph_dir: Optional[str] = None,
        missing_rate: float = 0.0,
        **kwargs,
    ):
        super().__init__(**kwargs)
        self.feat = feat
        self.gtype = gtype
        self.spec = parse_limits(feat)
        self.graph_dir = graph_dir
        self._cache: Dict[int, Optional[BatchedCFG]] = {}
        # Synthetic stand-in for the reference's missing_ids.txt (~7% of
        # examples whose CFG failed Joern parsing): deterministic per id.
        self.missing_rate = missing_rate

    # -- graph loading -------------------------------------------------------

    def _load_graph(self, _id: int) -> Optional[BatchedCFG]:
        if _id in self._cache:
            return self._cache[_id]
        g: Optional[BatchedCFG]
        if self.graph_dir is not None:
            path = os.path.join(self.graph_dir, f"{_id}.pt")
            g = BatchedCFG.load(path) if os.path.exists(path) else None
        else:
            gen = torch.Generator().manual_seed(int(_id) * 2654435761 % (2**31))
            if self.missing_rate > 0 and torch.rand((), generator=gen).item() < self.missing_rate:
                g = None
            else:
                row = self.df[self.df.id == _id]
                vul = int(row.vul.item()) if len(row) else 0
                g = synthetic_cfg(gen, input_dim=self.spec.input_dim, vuln_rate=0.0)
                if vul:
                    # mark a contiguous vulnerable region (approximates the
                    # line-level labels of the real pipeline)
                    n = g.num_nodes
                    k = max(1, n // 10)
                    start = int(torch.randint(0, max(1, n - k), (1,), generator=gen).item())
                    v = torch.zeros(n, dtype=torch.int64)
                    v[start : start + k] = 1
                    g.ndata["_VULN"] = v
        self._cache[_id] = g
        return g

    def item(self, idx: int) -> Tuple[BatchedCFG, Dict]:
        _id = self.idx2id[idx]
        g = self._load_graph(_id)
        if g is None:
            raise KeyError(f"graph for id {_id} is missing")
        return g, {}

    def __getitem__(self, idx: int):
        return self.item(int(idx))

    def get_indices(self, ids: Sequence[int]) -> Tuple[Optional[BatchedCFG], List[int]]:
        """Batch the CFGs for the given EXAMPLE IDS; returns (batched graph,
        positions-of-missing). Mirrors dataset.py:63-76: combined models drop
        text examples whose graph is missing."""
        graphs, missing = [], []
        for pos, _id in enumerate(ids):
            g = self._load_graph(int(_id))
            if g is None:
                missing.append(pos)
            else:
                graphs.append(g)
        if not graphs:
            return None, missing
        return batch_graphs(graphs), missing


def collate_graphs(items: List[Tuple[BatchedCFG, Dict]]) -> Tuple[BatchedCFG, Dict]:
    graphs = [g for g, _ in items]
    return batch_graphs(graphs), {}
