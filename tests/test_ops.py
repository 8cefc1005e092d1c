"""Op-level tests: our autograd ops vs independent torch-autograd oracles."""

import torch

from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
from deepdfa_amd.ops import attn_pool, embed4, gru_cell, segment_max, spmm_sum


def _dense_A(g):
    N = g.num_nodes
    A = torch.zeros(N, N, dtype=torch.float64)
    ip = g.indptr.tolist()
    idx = g.indices.tolist()
    for v in range(N):
        for e in range(ip[v], ip[v + 1]):
            A[v, idx[e]] += 1
    return A


def test_spmm_matches_dense_and_grads():
    g = synthetic_cfg_batch(4, seed=0)
    N = g.num_nodes
    x = torch.randn(N, 16, dtype=torch.float64, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    A = _dense_A(g)
    out_ref = A @ x2
    out = spmm_sum(x, g)
    assert torch.allclose(out.double(), out_ref, atol=1e-9)
    go = torch.randn_like(out_ref)
    out_ref.backward(go)
    out.backward(go.to(out.dtype))
    assert torch.allclose(x.grad.double(), x2.grad, atol=1e-9)


def test_embed4_matches_embedding_and_grads():
    V, D = 50, 8
    N = 40
    tables = torch.randn(4, V, D, requires_grad=True)
    idx = torch.randint(0, V, (N, 4))
    out = embed4(tables, idx)
    # oracle via nn.functional.embedding
    t2 = tables.detach().clone().requires_grad_(True)
    out_ref = torch.cat([torch.nn.functional.embedding(idx[:, i], t2[i]) for i in range(4)], 1)
    assert torch.allclose(out, out_ref)
    go = torch.randn_like(out)
    out.backward(go)
    out_ref.backward(go)
    assert torch.allclose(tables.grad, t2.grad, atol=1e-6)


def test_gru_cell_matches_torch_grucell():
    torch.manual_seed(0)
    N, H = 33, 24
    cell = torch.nn.GRUCell(H, H).double()
    a = torch.randn(N, H, dtype=torch.float64, requires_grad=True)
    h = torch.randn(N, H, dtype=torch.float64, requires_grad=True)
    a2 = a.detach().clone().requires_grad_(True)
    h2 = h.detach().clone().requires_grad_(True)
    out = gru_cell(a, h, cell.weight_ih, cell.weight_hh, cell.bias_ih, cell.bias_hh)
    out_ref = cell(a2, h2)
    assert torch.allclose(out, out_ref, atol=1e-10)
    go = torch.randn_like(out)
    out.backward(go)
    out_ref.backward(go)
    assert torch.allclose(a.grad, a2.grad, atol=1e-10)
    assert torch.allclose(h.grad, h2.grad, atol=1e-10)
    # weight grads flow through the matmuls
    assert cell.weight_ih.grad is not None and cell.weight_ih.grad.abs().sum() > 0


def test_gru_cell_gradcheck():
    N, H = 5, 6
    cell = torch.nn.GRUCell(H, H).double()
    a = torch.randn(N, H, dtype=torch.float64, requires_grad=True)
    h = torch.randn(N, H, dtype=torch.float64, requires_grad=True)

    def f(a_, h_):
        return gru_cell(a_, h_, cell.weight_ih, cell.weight_hh, cell.bias_ih, cell.bias_hh)

    assert torch.autograd.gradcheck(f, (a, h), eps=1e-6, atol=1e-8)


def test_attn_pool_matches_softmax_oracle():
    g = synthetic_cfg_batch(5, seed=2)
    N, B = g.num_nodes, g.num_graphs
    D = 12
    x = torch.randn(N, D, dtype=torch.float64, requires_grad=True)
    gate = torch.randn(N, dtype=torch.float64, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    gate2 = gate.detach().clone().requires_grad_(True)
    out = attn_pool(x, gate, g)
    # oracle: per-graph softmax + weighted sum via torch autograd
    offs = g.node_offsets.tolist()
    outs = []
    for b in range(B):
        lo, hi = offs[b], offs[b + 1]
        alpha = torch.softmax(gate2[lo:hi], dim=0)
        outs.append((alpha.unsqueeze(1) * x2[lo:hi]).sum(0))
    out_ref = torch.stack(outs)
    assert torch.allclose(out, out_ref, atol=1e-9)
    go = torch.randn_like(out_ref)
    out.backward(go)
    out_ref.backward(go)
    assert torch.allclose(x.grad, x2.grad, atol=1e-9)
    assert torch.allclose(gate.grad, gate2.grad, atol=1e-9)


def test_attn_pool_gradcheck():
    g = synthetic_cfg_batch(3, seed=9)
    N = g.num_nodes
    x = torch.randn(N, 4, dtype=torch.float64, requires_grad=True)
    gate = torch.randn(N, dtype=torch.float64, requires_grad=True)

    def f(x_, gate_):
        return attn_pool(x_, gate_, g)

    assert torch.autograd.gradcheck(f, (x, gate), eps=1e-6, atol=1e-8)


def test_segment_max_labels():
    g = synthetic_cfg_batch(6, seed=4)
    lab = segment_max(g.ndata["_VULN"].float(), g)
    offs = g.node_offsets.tolist()
    ref = torch.stack(
        [g.ndata["_VULN"][offs[b] : offs[b + 1]].max() for b in range(g.num_graphs)]
    ).float()
    assert torch.equal(lab, ref)
