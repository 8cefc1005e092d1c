"""Batched control-flow graphs as block-diagonal CSR/CSC.

MI355X-native replacement for DGL's batched graph (reference uses
`dgl.batch` + `DGLGraph`; see /root/reference/DDFA/sastvd/linevd/dataset.py:47
and graphmogrifier.py:59-97). A batch of B CFGs is ONE graph whose adjacency
is block-diagonal:

  * in-CSR  (indptr/indices over destination nodes) drives the forward
    message aggregation  m_v = sum_{u->v} x_u   — one contiguous
    segment-sum kernel over the whole batch, no per-graph launches;
  * out-CSC (t_indptr/t_indices) is the exact transpose, used by the
    backward pass  grad_x_u = sum_{v: u->v} grad_m_v;
  * node_offsets give per-graph node segments for the pooling /
    label-reduction kernels (segment softmax, segment max).

All index arrays are int32 (N, E < 2^31 always holds for these datasets) to
halve index bandwidth on the gather kernels; ndata stays int64/float as the
reference's formats demand.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Sequence, Tuple

import torch

__all__ = ["BatchedCFG", "batch_graphs"]


def _build_csr(
    num_nodes: int, src: torch.Tensor, dst: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """CSR over `dst`: indptr[v]..indptr[v+1] lists the source of every edge
    into v. Deterministic (stable sort by dst, ties keep input order)."""
    counts = torch.bincount(dst, minlength=num_nodes)
    indptr = torch.zeros(num_nodes + 1, dtype=torch.int64)
    torch.cumsum(counts, 0, out=indptr[1:])
    order = torch.argsort(dst, stable=True)
    indices = src[order].contiguous()
    return indptr.to(torch.int32), indices.to(torch.int32)


class BatchedCFG:
    """One (possibly batched) CFG. Construct via from_edges / batch_graphs."""

    def __init__(
        self,
        node_offsets: torch.Tensor,  # (B+1,) int32
        indptr: torch.Tensor,  # (N+1,) int32  in-edge CSR
        indices: torch.Tensor,  # (E,)  int32  source node per in-edge
        t_indptr: torch.Tensor,  # (N+1,) int32  out-edge CSC (transpose)
        t_indices: torch.Tensor,  # (E,)  int32
        ndata: Optional[Dict[str, torch.Tensor]] = None,
    ):
        self.node_offsets = node_offsets
        self.indptr = indptr
        self.indices = indices
        self.t_indptr = t_indptr
        self.t_indices = t_indices
        self.ndata: Dict[str, torch.Tensor] = ndata if ndata is not None else {}
        self._segment_ids: Optional[torch.Tensor] = None

    # -- construction -------------------------------------------------------

    @staticmethod
    def from_edges(
        num_nodes: int,
        src: Sequence[int],
        dst: Sequence[int],
        ndata: Optional[Dict[str, torch.Tensor]] = None,
        add_self_loops: bool = True,
    ) -> "BatchedCFG":
        """Build a single-graph batch. Self-loop insertion mirrors the
        reference's dbize_graphs.py:25 (`add_self_loop` on every node, after
        removing any duplicate self-loops present in the edge list)."""
        src_t = torch.as_tensor(src, dtype=torch.int64)
        dst_t = torch.as_tensor(dst, dtype=torch.int64)
        if src_t.numel():
            if int(src_t.max()) >= num_nodes or int(dst_t.max()) >= num_nodes:
                raise ValueError("edge endpoint out of range")
        if add_self_loops:
            keep = src_t != dst_t
            src_t, dst_t = src_t[keep], dst_t[keep]
            loop = torch.arange(num_nodes, dtype=torch.int64)
            src_t = torch.cat([src_t, loop])
            dst_t = torch.cat([dst_t, loop])
        indptr, indices = _build_csr(num_nodes, src_t, dst_t)
        t_indptr, t_indices = _build_csr(num_nodes, dst_t, src_t)
        node_offsets = torch.tensor([0, num_nodes], dtype=torch.int32)
        return BatchedCFG(node_offsets, indptr, indices, t_indptr, t_indices, ndata)

    # -- shape accessors -----------------------------------------------------

    @property
    def num_graphs(self) -> int:
        return self.node_offsets.numel() - 1

    @property
    def num_nodes(self) -> int:
        return self.indptr.numel() - 1

    @property
    def num_edges(self) -> int:
        return self.indices.numel()

    def batch_num_nodes(self) -> torch.Tensor:
        return (self.node_offsets[1:] - self.node_offsets[:-1]).to(torch.int64)

    def segment_ids(self) -> torch.Tensor:
        """(N,) int32 graph id per node (cached)."""
        if self._segment_ids is None or self._segment_ids.numel() != self.num_nodes:
            counts = (self.node_offsets[1:] - self.node_offsets[:-1]).to(torch.int64)
            self._segment_ids = torch.repeat_interleave(
                torch.arange(self.num_graphs, dtype=torch.int32, device=counts.device),
                counts,
            )
        return self._segment_ids

    # -- device movement -----------------------------------------------------

    def to(self, device, non_blocking: bool = False) -> "BatchedCFG":
        g = BatchedCFG(
            self.node_offsets.to(device, non_blocking=non_blocking),
            self.indptr.to(device, non_blocking=non_blocking),
            self.indices.to(device, non_blocking=non_blocking),
            self.t_indptr.to(device, non_blocking=non_blocking),
            self.t_indices.to(device, non_blocking=non_blocking),
            {k: v.to(device, non_blocking=non_blocking) for k, v in self.ndata.items()},
        )
        if self._segment_ids is not None:
            g._segment_ids = self._segment_ids.to(device, non_blocking=non_blocking)
        return g

    @property
    def device(self):
        return self.indptr.device

    # -- (de)serialization ---------------------------------------------------

    def state(self) -> Dict[str, torch.Tensor]:
        d = {
            "node_offsets": self.node_offsets,
            "indptr": self.indptr,
            "indices": self.indices,
            "t_indptr": self.t_indptr,
            "t_indices": self.t_indices,
        }
        for k, v in self.ndata.items():
            d["ndata:" + k] = v
        return d

    @staticmethod
    def from_state(d: Dict[str, torch.Tensor]) -> "BatchedCFG":
        ndata = {k[6:]: v for k, v in d.items() if k.startswith("ndata:")}
        return BatchedCFG(
            d["node_offsets"], d["indptr"], d["indices"], d["t_indptr"], d["t_indices"], ndata
        )

    def save(self, path: str) -> None:
        torch.save(self.state(), path)

    @staticmethod
    def load(path: str) -> "BatchedCFG":
        return BatchedCFG.from_state(torch.load(path, weights_only=True))

    # -- unbatch (rarely needed; label reduction is a kernel) ---------------

    def unbatch(self) -> List["BatchedCFG"]:
        out = []
        offs = self.node_offsets.tolist()
        for g in range(self.num_graphs):
            lo, hi = offs[g], offs[g + 1]
            n = hi - lo
            # slice in-CSR rows [lo, hi)
            ip = (self.indptr[lo : hi + 1].to(torch.int64) - int(self.indptr[lo])).to(torch.int32)
            e_lo, e_hi = int(self.indptr[lo]), int(self.indptr[hi])
            idx = (self.indices[e_lo:e_hi].to(torch.int64) - lo).to(torch.int32)
            tp = (self.t_indptr[lo : hi + 1].to(torch.int64) - int(self.t_indptr[lo])).to(
                torch.int32
            )
            te_lo, te_hi = int(self.t_indptr[lo]), int(self.t_indptr[hi])
            tidx = (self.t_indices[te_lo:te_hi].to(torch.int64) - lo).to(torch.int32)
            nd = {k: v[lo:hi] for k, v in self.ndata.items()}
            out.append(
                BatchedCFG(torch.tensor([0, n], dtype=torch.int32), ip, idx, tp, tidx, nd)
            )
        return out


def batch_graphs(graphs: List[BatchedCFG]) -> BatchedCFG:
    """Concatenate graphs into one block-diagonal batch (dgl.batch analog)."""
    if len(graphs) == 1:
        return graphs[0]
    node_off = [0]
    edge_off_ip: List[torch.Tensor] = []
    edge_off_tip: List[torch.Tensor] = []
    idxs: List[torch.Tensor] = []
    tidxs: List[torch.Tensor] = []
    e_base = 0
    te_base = 0
    for g in graphs:
        n_base = node_off[-1]
        for off in (g.node_offsets[1:].to(torch.int64) + n_base).tolist():
            node_off.append(off)
        edge_off_ip.append(g.indptr[:-1].to(torch.int64) + e_base)
        edge_off_tip.append(g.t_indptr[:-1].to(torch.int64) + te_base)
        idxs.append(g.indices.to(torch.int64) + n_base)
        tidxs.append(g.t_indices.to(torch.int64) + n_base)
        e_base += g.num_edges
        te_base += g.num_edges
    N = node_off[-1]
    indptr = torch.cat(edge_off_ip + [torch.tensor([e_base], dtype=torch.int64)]).to(torch.int32)
    t_indptr = torch.cat(edge_off_tip + [torch.tensor([te_base], dtype=torch.int64)]).to(
        torch.int32
    )
    indices = torch.cat(idxs).to(torch.int32)
    t_indices = torch.cat(tidxs).to(torch.int32)
    node_offsets = torch.tensor(node_off, dtype=torch.int32)
    keys = graphs[0].ndata.keys()
    ndata = {k: torch.cat([g.ndata[k] for g in graphs]) for k in keys}
    assert indptr.numel() == N + 1
    return BatchedCFG(node_offsets, indptr, indices, t_indptr, t_indices, ndata)
