"""RoBERTa encoder parity vs HuggingFace transformers (CPU oracle)."""

import pytest
import torch

from deepdfa_amd.models.linevul import Model, RobertaClassificationHead
from deepdfa_amd.models.roberta import RobertaConfig, RobertaModel

transformers = pytest.importorskip("transformers")


def small_cfg(layers=2):
    return RobertaConfig(
        vocab_size=200,
        hidden_size=64,
        num_hidden_layers=layers,
        num_attention_heads=4,
        intermediate_size=128,
        max_position_embeddings=66,
    )


def hf_cfg(cfg):
    return transformers.RobertaConfig(
        vocab_size=cfg.vocab_size,
        hidden_size=cfg.hidden_size,
        num_hidden_layers=cfg.num_hidden_layers,
        num_attention_heads=cfg.num_attention_heads,
        intermediate_size=cfg.intermediate_size,
        max_position_embeddings=cfg.max_position_embeddings,
        type_vocab_size=cfg.type_vocab_size,
        hidden_act="gelu",
        layer_norm_eps=cfg.layer_norm_eps,
        attn_implementation="eager",
    )


def make_pair(cfg):
    torch.manual_seed(0)
    hf = transformers.RobertaModel(hf_cfg(cfg), add_pooling_layer=False)
    ours = RobertaModel(cfg)
    missing, unexpected = ours.load_state_dict(hf.state_dict(), strict=False)
    assert not missing, missing
    assert not unexpected, unexpected
    return hf.eval(), ours.eval()


def padded_ids(B=3, L=24, vocab=200, seed=1):
    torch.manual_seed(seed)
    ids = torch.randint(3, vocab, (B, L))
    ids[0, 18:] = 1  # suffix padding (pad_token_id = 1)
    ids[2, 10:] = 1
    return ids


def test_state_dict_keys_match_hf():
    cfg = small_cfg()
    hf = transformers.RobertaModel(hf_cfg(cfg), add_pooling_layer=False)
    ours = RobertaModel(cfg)
    assert set(hf.state_dict().keys()) == set(ours.state_dict().keys())


def test_forward_parity_with_hf():
    cfg = small_cfg()
    hf, ours = make_pair(cfg)
    ids = padded_ids()
    mask = ids.ne(1)
    with torch.no_grad():
        ref = hf(ids, attention_mask=mask).last_hidden_state
        out, _ = ours(ids, attention_mask=mask)
    # compare only non-pad positions (HF computes garbage on pads too but
    # identical; our masked softmax zeroes pad keys the same way)
    m = mask.unsqueeze(-1)
    diff = ((ref - out) * m).abs().max().item()
    assert diff < 2e-4, diff


def test_attention_probs_parity():
    cfg = small_cfg(layers=1)
    hf, ours = make_pair(cfg)
    ids = padded_ids()
    mask = ids.ne(1)
    with torch.no_grad():
        ref = hf(ids, attention_mask=mask, output_attentions=True).attentions[0]
        _, probs = ours(ids, attention_mask=mask, output_attentions=True)
    valid_rows = mask.view(3, 1, -1, 1)
    diff = ((ref - probs[0]) * valid_rows).abs().max().item()
    assert diff < 2e-4, diff


def test_backward_parity_with_hf():
    cfg = small_cfg(layers=1)
    hf, ours = make_pair(cfg)
    hf.train(False)
    ours.train(False)
    ids = padded_ids()
    mask = ids.ne(1)
    ref = hf(ids, attention_mask=mask).last_hidden_state
    (ref[mask].square().mean()).backward()
    out, _ = ours(ids, attention_mask=mask)
    (out[mask].square().mean()).backward()
    hf_grads = {k: v.grad for k, v in hf.named_parameters()}
    for name, p in ours.named_parameters():
        if p.grad is None:
            continue
        g_ref = hf_grads[name]
        assert g_ref is not None, name
        diff = (p.grad - g_ref).abs().max()
        assert diff < 1e-6 + 5e-3 * g_ref.abs().max(), (name, diff.item())


def test_linevul_model_forward():
    cfg = small_cfg()
    torch.manual_seed(0)
    model = Model(config=cfg)
    ids = padded_ids()
    labels = torch.tensor([0, 1, 0])
    loss, prob = model(ids, labels=labels)
    assert prob.shape == (3, 2)
    assert torch.isfinite(loss)
    loss.backward()


def test_linevul_combined_with_flowgnn():
    from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
    from deepdfa_amd.models import FlowGNNGGNNModule

    cfg = small_cfg()
    torch.manual_seed(0)
    fg = FlowGNNGGNNModule(input_dim=1002, hidden_dim=32, n_steps=2, num_output_layers=3,
                           encoder_mode=True)
    model = Model(config=cfg, flowgnn_encoder=fg)
    assert model.classifier.dense.in_features == cfg.hidden_size + 256
    ids = padded_ids()
    g = synthetic_cfg_batch(3, seed=0)
    labels = torch.tensor([0, 1, 1])
    loss, prob = model(ids, labels=labels, graphs=g)
    assert torch.isfinite(loss)
    loss.backward()
    # flow-GNN grads flow through the combined head
    assert fg.ggnn.gru.weight_ih.grad is not None
