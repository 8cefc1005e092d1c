"""FlowGNN model tests: shape, determinism, full-model parity vs a
torch-native oracle (nn.GRUCell + dense adjacency), encoder mode."""

import torch

from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
from deepdfa_amd.models import FlowGNNGGNNModule

FEAT = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000"


def build_model(**kw):
    torch.manual_seed(0)
    return FlowGNNGGNNModule(feat=FEAT, input_dim=1002, hidden_dim=32, n_steps=5,
                             num_output_layers=3, **kw)


def oracle_forward(model, g):
    """Independent re-implementation with torch primitives + dense adjacency."""
    feats = ["api", "datatype", "literal", "operator"]
    embeds = [model.all_embeddings[f](g.ndata[f"_ABS_DATAFLOW_{f}"]) for f in feats]
    x = torch.cat(embeds, dim=1)
    N = g.num_nodes
    A = torch.zeros(N, N)
    ip, idx = g.indptr.tolist(), g.indices.tolist()
    for v in range(N):
        for e in range(ip[v], ip[v + 1]):
            A[v, idx[e]] += 1
    h = x
    cell = model.ggnn.gru
    for _ in range(model.ggnn.n_steps):
        wh = model.ggnn.linear(h)
        m = A @ wh
        h = cell(m, h)
    out = torch.cat([h, x], dim=1)
    offs = g.node_offsets.tolist()
    pooled = []
    gate = model.pooling.gate_nn(out).squeeze(-1)
    for b in range(g.num_graphs):
        lo, hi = offs[b], offs[b + 1]
        alpha = torch.softmax(gate[lo:hi], 0)
        pooled.append((alpha.unsqueeze(1) * out[lo:hi]).sum(0))
    pooled = torch.stack(pooled)
    return model.output_layer(pooled).squeeze(-1)


def test_forward_shapes():
    model = build_model()
    g = synthetic_cfg_batch(8, seed=0)
    logits = model(g, {})
    assert logits.shape == (8,)


def test_forward_matches_oracle():
    model = build_model()
    model.eval()
    g = synthetic_cfg_batch(6, seed=1)
    with torch.no_grad():
        ours = model(g, {})
        ref = oracle_forward(model, g)
    assert torch.allclose(ours, ref, atol=1e-4), (ours - ref).abs().max()


def test_backward_matches_oracle():
    model = build_model()
    g = synthetic_cfg_batch(4, seed=2)
    ours = model(g, {}).sum()
    ours.backward()
    grads = {n: p.grad.clone() for n, p in model.named_parameters() if p.grad is not None}
    model.zero_grad()
    ref = oracle_forward(model, g).sum()
    ref.backward()
    for n, p in model.named_parameters():
        if p.grad is None:
            continue
        assert n in grads
        assert torch.allclose(grads[n], p.grad, atol=1e-3), (n, (grads[n] - p.grad).abs().max())


def test_encoder_mode():
    model = build_model(encoder_mode=True)
    g = synthetic_cfg_batch(5, seed=3)
    emb = model(g, {})
    assert emb.shape == (5, 256)
    assert model.out_dim == 256


def test_training_step_and_label():
    model = build_model(positive_weight=2.0)
    g = synthetic_cfg_batch(8, seed=4)
    loss = model.training_step((g, {}))
    assert loss.requires_grad and loss.ndim == 0
    loss.backward()
    m = model.epoch_metrics("train")
    assert "train_f1" in m


def test_state_dict_key_names():
    """Checkpoint-layout compatibility: reference key names."""
    model = build_model()
    keys = set(model.state_dict().keys())
    for expect in [
        "all_embeddings.api.weight",
        "all_embeddings.operator.weight",
        "ggnn.linear.weight",
        "ggnn.gru.weight_ih",
        "ggnn.gru.bias_hh",
        "pooling.gate_nn.weight",
        "output_layer.0.weight",
        "output_layer.2.weight",
        "output_layer.4.weight",
    ]:
        assert expect in keys, expect


def test_determinism_same_seed():
    m1 = build_model()
    m2 = build_model()
    g = synthetic_cfg_batch(3, seed=7)
    with torch.no_grad():
        assert torch.equal(m1(g, {}), m2(g, {}))
