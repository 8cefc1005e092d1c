"""Pad a BatchedCFG to fixed (nodes, edges, graphs) shape buckets.

hipGraph capture requires static tensor shapes across replays; Big-Vul CFG
batches vary in total node/edge count every step. pad_batch() appends dummy
graphs that absorb the padding so a captured step can be replayed on any
batch of the same bucket:

  * padded nodes live ONLY in appended dummy graphs (real graphs untouched);
    their features are 0 ("not a definition" per dbize_absdf.py:35-42) and
    _VULN = 0;
  * padded edges are self-loops on dummy nodes (parallel self-loops on the
    last dummy node absorb any excess) — block-diagonal structure and all
    segment kernels stay valid;
  * every appended graph has >= 1 node, so segment softmax/max never see an
    empty segment;
  * the returned weight vector is 1 for real graphs, 0 for dummies — the
    loss and metric updates mask them out exactly (see
    BaseModule.loss_fn(weight=...) / BinaryStats.update(mask=...)).
"""

from __future__ import annotations

from typing import Tuple

import torch

from .batch import BatchedCFG

__all__ = ["pad_batch", "bucket_shape"]


def bucket_shape(
    g: BatchedCFG, b_pad: int, node_q: int = 1024, edge_q: int = 4096
) -> Tuple[int, int, int]:
    """Quantized target shape for a batch: graphs fixed at b_pad, nodes and
    edges rounded up to the next quantum (with room for the dummy graphs'
    mandatory 1 node / 1 self-loop each)."""
    b_extra = b_pad - g.num_graphs
    assert b_extra >= 1, "b_pad must exceed the real graph count"
    n_min = g.num_nodes + b_extra
    n_pad = ((n_min + node_q - 1) // node_q) * node_q
    # every padded node carries >= 1 self-loop, so the edge budget must
    # cover the NODE padding (not just the dummy-graph minimum)
    e_min = g.num_edges + (n_pad - g.num_nodes)
    e_pad = ((e_min + edge_q - 1) // edge_q) * edge_q
    return n_pad, e_pad, b_pad


def pad_batch(
    g: BatchedCFG, n_pad: int, e_pad: int, b_pad: int
) -> Tuple[BatchedCFG, torch.Tensor]:
    """Return (padded BatchedCFG with exactly (n_pad, e_pad, b_pad),
    graph weight vector (b_pad,) float32 with 1=real / 0=dummy)."""
    N, E, B = g.num_nodes, g.num_edges, g.num_graphs
    b_extra = b_pad - B
    n_extra = n_pad - N
    e_extra = e_pad - E
    assert b_extra >= 1 and n_extra >= b_extra and e_extra >= n_extra, (
        f"padding target too small: ({n_pad},{e_pad},{b_pad}) for ({N},{E},{B})"
    )
    dev = g.indptr.device
    # dummy graph 0 absorbs the surplus nodes; the rest get 1 node each
    nodes_per_dummy = [n_extra - (b_extra - 1)] + [1] * (b_extra - 1)
    node_offsets = torch.cat(
        [
            g.node_offsets.to(dev).to(torch.int64),
            N + torch.cumsum(torch.tensor(nodes_per_dummy, dtype=torch.int64, device=dev), 0),
        ]
    ).to(torch.int32)

    # self-loop edges: one per dummy node, surplus as parallel self-loops on
    # the LAST dummy node (so CSR row pointers stay monotone)
    base_loops = n_extra
    surplus = e_extra - base_loops
    counts = torch.ones(n_extra, dtype=torch.int64, device=dev)
    counts[-1] += surplus
    pad_indptr = g.indptr[-1].to(torch.int64) + torch.cumsum(counts, 0)
    loop_src = torch.repeat_interleave(
        torch.arange(N, N + n_extra, dtype=torch.int64, device=dev), counts
    ).to(torch.int32)
    indptr = torch.cat([g.indptr.to(torch.int64), pad_indptr]).to(torch.int32)
    indices = torch.cat([g.indices, loop_src])
    # the transpose of a self-loop block is itself
    t_pad_indptr = g.t_indptr[-1].to(torch.int64) + torch.cumsum(counts, 0)
    t_indptr = torch.cat([g.t_indptr.to(torch.int64), t_pad_indptr]).to(torch.int32)
    t_indices = torch.cat([g.t_indices, loop_src])

    ndata = {}
    for k, v in g.ndata.items():
        pad_shape = (n_extra,) + tuple(v.shape[1:])
        ndata[k] = torch.cat([v, torch.zeros(pad_shape, dtype=v.dtype, device=v.device)])

    weight = torch.zeros(b_pad, dtype=torch.float32, device=dev)
    weight[:B] = 1.0
    out = BatchedCFG(node_offsets, indptr, indices, t_indptr, t_indices, ndata)
    assert out.num_nodes == n_pad and out.num_edges == e_pad and out.num_graphs == b_pad
    return out, weight
