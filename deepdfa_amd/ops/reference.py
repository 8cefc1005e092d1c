"""Pure-PyTorch fp32 reference implementations of every custom op.

These are the correctness oracles for the HIP/CDNA4 kernels (SURVEY.md §4:
"Numerics tests for a HIP kernel compare it against a plain PyTorch fp32
reference of the same op") and the CPU execution path. Shapes follow the
reference model (ggnn.py:22-109): N nodes, D=128 hidden, B graphs.
"""

from __future__ import annotations

from typing import List, Tuple

import torch


# -- K1: 4-way embedding gather-concat --------------------------------------

def embed4_fwd(tables: List[torch.Tensor], idx: torch.Tensor) -> torch.Tensor:
    """tables: 4 x (V, Demb); idx: (N, 4) int64 -> (N, 4*Demb).
    Mirrors ggnn.py:84-89 (per-feature nn.Embedding lookup then concat)."""
    return torch.cat([t[idx[:, i]] for i, t in enumerate(tables)], dim=1)


def embed4_bwd(
    grad_out: torch.Tensor, idx: torch.Tensor, vocab: int, demb: int
) -> List[torch.Tensor]:
    grads = []
    for i in range(idx.shape[1]):
        g = torch.zeros(vocab, demb, dtype=grad_out.dtype, device=grad_out.device)
        g.index_add_(0, idx[:, i], grad_out[:, i * demb : (i + 1) * demb])
        grads.append(g)
    return grads


# -- K2: CSR segment-sum (message aggregation) -------------------------------

def spmm_sum(indptr: torch.Tensor, indices: torch.Tensor, x: torch.Tensor) -> torch.Tensor:
    """m[v] = sum over in-edges (u -> v) of x[u]. indptr (N+1,), indices (E,).
    The DGL `update_all(copy_u, sum)` equivalent for one etype."""
    N = indptr.numel() - 1
    deg = (indptr[1:] - indptr[:-1]).to(torch.int64)
    seg = torch.repeat_interleave(torch.arange(N, device=x.device), deg)
    out = torch.zeros(N, x.shape[1], dtype=x.dtype, device=x.device)
    out.index_add_(0, seg, x[indices.to(torch.int64)])
    return out


# -- K3: GRU cell gates ------------------------------------------------------

def gru_gates_fwd(
    gi: torch.Tensor, gh: torch.Tensor, h: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor, torch.Tensor]:
    """torch.nn.GRUCell semantics (DGL GatedGraphConv's internal GRUCell):
    gi = a @ W_ih^T + b_ih (N, 3H)  [input = aggregated message a]
    gh = h @ W_hh^T + b_hh (N, 3H)
    r = sigmoid(gi_r + gh_r); z = sigmoid(gi_z + gh_z)
    n = tanh(gi_n + r * gh_n); h' = (1 - z) * n + z * h
    Returns (h_new, r, z, n) — r/z/n saved for backward."""
    H = h.shape[1]
    i_r, i_z, i_n = gi[:, :H], gi[:, H : 2 * H], gi[:, 2 * H :]
    h_r, h_z, h_n = gh[:, :H], gh[:, H : 2 * H], gh[:, 2 * H :]
    r = torch.sigmoid(i_r + h_r)
    z = torch.sigmoid(i_z + h_z)
    n = torch.tanh(i_n + r * h_n)
    h_new = (1.0 - z) * n + z * h
    return h_new, r, z, n


def gru_gates_bwd(
    grad_h_new: torch.Tensor,
    gh: torch.Tensor,
    h: torch.Tensor,
    r: torch.Tensor,
    z: torch.Tensor,
    n: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """Returns (grad_gi, grad_gh, grad_h)."""
    H = h.shape[1]
    h_n = gh[:, 2 * H :]
    dn = grad_h_new * (1.0 - z)
    dz = grad_h_new * (h - n)
    d_preact_n = dn * (1.0 - n * n)  # tanh'
    dr = d_preact_n * h_n
    d_preact_r = dr * r * (1.0 - r)  # sigmoid'
    d_preact_z = dz * z * (1.0 - z)
    grad_gi = torch.cat([d_preact_r, d_preact_z, d_preact_n], dim=1)
    grad_gh = torch.cat([d_preact_r, d_preact_z, d_preact_n * r], dim=1)
    grad_h = grad_h_new * z
    return grad_gi, grad_gh, grad_h


# -- K5: gated attention pooling (segment softmax + weighted segment sum) ----

def attn_pool_fwd(
    x: torch.Tensor, gate_logits: torch.Tensor, node_offsets: torch.Tensor
) -> Tuple[torch.Tensor, torch.Tensor]:
    """DGL GlobalAttentionPooling equivalent (invoked at ggnn.py:102):
    alpha = segment_softmax(gate_logits); out[g] = sum_v alpha_v * x_v.
    x (N, D); gate_logits (N,); node_offsets (B+1,). Returns (out (B,D), alpha (N,))."""
    B = node_offsets.numel() - 1
    counts = (node_offsets[1:] - node_offsets[:-1]).to(torch.int64)
    seg = torch.repeat_interleave(torch.arange(B, device=x.device), counts)
    # segment softmax (max-subtracted, matching the kernel's numerics)
    neg_inf = torch.finfo(gate_logits.dtype).min
    seg_max = torch.full((B,), neg_inf, dtype=gate_logits.dtype, device=x.device)
    seg_max.scatter_reduce_(0, seg, gate_logits, reduce="amax", include_self=True)
    e = torch.exp(gate_logits - seg_max[seg])
    denom = torch.zeros(B, dtype=gate_logits.dtype, device=x.device)
    denom.index_add_(0, seg, e)
    alpha = e / denom[seg]
    out = torch.zeros(B, x.shape[1], dtype=x.dtype, device=x.device)
    out.index_add_(0, seg, x * alpha.unsqueeze(1))
    return out, alpha


def attn_pool_bwd(
    grad_out: torch.Tensor,
    x: torch.Tensor,
    alpha: torch.Tensor,
    node_offsets: torch.Tensor,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """Returns (grad_x (N,D), grad_gate_logits (N,)).
    d out[g] / d x_v = alpha_v ; softmax backward for the gate."""
    B = node_offsets.numel() - 1
    counts = (node_offsets[1:] - node_offsets[:-1]).to(torch.int64)
    seg = torch.repeat_interleave(torch.arange(B, device=x.device), counts)
    go = grad_out[seg]  # (N, D)
    grad_x = go * alpha.unsqueeze(1)
    # s_v = <grad_out[g], x_v>; grad_gate = alpha * (s - sum_g alpha*s)
    s = (go * x).sum(dim=1)
    dot = torch.zeros(B, dtype=x.dtype, device=x.device)
    dot.index_add_(0, seg, alpha * s)
    grad_gate = alpha * (s - dot[seg])
    return grad_x, grad_gate


# -- K7: per-graph label max -------------------------------------------------

def segment_max(values: torch.Tensor, node_offsets: torch.Tensor) -> torch.Tensor:
    """Graph label = max of node labels (base_module.py:83-95 get_label)."""
    B = node_offsets.numel() - 1
    counts = (node_offsets[1:] - node_offsets[:-1]).to(torch.int64)
    seg = torch.repeat_interleave(torch.arange(B, device=values.device), counts)
    out = torch.zeros(B, dtype=values.dtype, device=values.device)
    out.scatter_reduce_(0, seg, values, reduce="amax", include_self=False)
    return out
