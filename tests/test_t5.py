"""T5 stack parity vs HuggingFace transformers (CPU oracle) + DefectModel."""

import pytest
import torch

from deepdfa_amd.models.codet5 import DefectModel
from deepdfa_amd.models.t5 import T5Config, T5ForConditionalGeneration

transformers = pytest.importorskip("transformers")


def small_cfg(layers=2):
    return T5Config(
        vocab_size=300, d_model=64, d_kv=16, d_ff=128, num_layers=layers,
        num_decoder_layers=layers, num_heads=4,
    )


def hf_model(cfg):
    hc = transformers.T5Config(
        vocab_size=cfg.vocab_size, d_model=cfg.d_model, d_kv=cfg.d_kv, d_ff=cfg.d_ff,
        num_layers=cfg.num_layers, num_decoder_layers=cfg.num_decoder_layers,
        num_heads=cfg.num_heads, dropout_rate=cfg.dropout_rate,
        decoder_start_token_id=cfg.decoder_start_token_id,
        feed_forward_proj="relu", tie_word_embeddings=True,
        attn_implementation="eager",
    )
    return transformers.T5ForConditionalGeneration(hc)


def make_pair(cfg):
    torch.manual_seed(0)
    hf = hf_model(cfg)
    ours = T5ForConditionalGeneration(cfg)
    missing, unexpected = ours.load_state_dict(hf.state_dict(), strict=False)
    assert not missing, missing
    return hf.eval(), ours.eval()


def padded_ids(B=3, L=20, vocab=300, seed=2):
    torch.manual_seed(seed)
    ids = torch.randint(3, vocab, (B, L))
    ids[0, 14:] = 0
    ids[2, 9:] = 0
    # EOS (id 2) at the last non-pad position of each row
    ids[0, 13] = 2
    ids[1, L - 1] = 2
    ids[2, 8] = 2
    return ids


def test_state_dict_keys_match_hf():
    cfg = small_cfg()
    hf = hf_model(cfg)
    ours = T5ForConditionalGeneration(cfg)
    hf_keys = set(hf.state_dict().keys())
    our_keys = set(ours.state_dict().keys())
    assert hf_keys - our_keys == set(), hf_keys - our_keys


def test_seq2seq_logits_parity():
    cfg = small_cfg()
    hf, ours = make_pair(cfg)
    ids = padded_ids()
    mask = ids.ne(0)
    with torch.no_grad():
        ref = hf(input_ids=ids, attention_mask=mask, labels=ids)
        loss, logits, _dec = ours(ids, attention_mask=mask, labels=ids)
    diff = (ref.logits - logits).abs().max().item()
    assert diff < 3e-4, diff
    assert abs(ref.loss.item() - loss.item()) < 1e-3


def test_seq2seq_backward_parity():
    cfg = small_cfg(layers=1)
    hf, ours = make_pair(cfg)
    ids = padded_ids()
    mask = ids.ne(0)
    hf(input_ids=ids, attention_mask=mask, labels=ids).loss.backward()
    loss, _, _ = ours(ids, attention_mask=mask, labels=ids)
    loss.backward()
    hfg = {k: v.grad for k, v in hf.named_parameters()}
    for name, p in ours.named_parameters():
        if p.grad is None:
            continue
        g_ref = hfg.get(name)
        if g_ref is None:
            continue
        diff = (p.grad - g_ref).abs().max()
        assert diff < 1e-5 + 5e-3 * g_ref.abs().max(), (name, diff.item())


def test_defect_model_forward_backward():
    cfg = small_cfg()
    torch.manual_seed(0)
    model = DefectModel(config=cfg)
    ids = padded_ids()
    labels = torch.tensor([0, 1, 1])
    loss, prob = model(ids, labels=labels)
    assert prob.shape == (3, 2)
    assert torch.isfinite(loss)
    loss.backward()


def test_defect_model_combined_flowgnn():
    from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
    from deepdfa_amd.models import FlowGNNGGNNModule

    cfg = small_cfg()
    torch.manual_seed(0)
    fg = FlowGNNGGNNModule(input_dim=1002, hidden_dim=32, n_steps=2,
                           num_output_layers=3, encoder_mode=True)
    model = DefectModel(config=cfg, flowgnn_encoder=fg)
    assert model.classifier.in_features == cfg.d_model + 256
    ids = padded_ids()
    g = synthetic_cfg_batch(3, seed=1)
    loss, prob = model(ids, labels=torch.tensor([1, 0, 1]), graphs=g)
    assert torch.isfinite(loss)
    loss.backward()
    assert fg.ggnn.gru.weight_ih.grad is not None


def test_eos_count_check():
    cfg = small_cfg()
    model = DefectModel(config=cfg)
    ids = padded_ids()
    ids[0, 5] = 2  # extra EOS in row 0 only
    with pytest.raises(ValueError):
        model(ids, labels=torch.tensor([0, 1, 0]))


class TestKVCachedGeneration:
    """generate() now decodes incrementally with per-layer KV caches; it
    must produce exactly the tokens of a full-prefix re-run per step."""

    def _model(self):
        torch.manual_seed(3)
        cfg = T5Config(num_layers=2, num_decoder_layers=2, d_model=64,
                       d_ff=128, num_heads=2, vocab_size=200)
        m = T5ForConditionalGeneration(cfg).eval()
        return m, cfg

    def _reference_greedy(self, m, cfg, ids, max_length):
        # the old quadratic scheme: full decoder re-run per step
        attention_mask = ids.ne(cfg.pad_token_id)
        enc_valid = attention_mask.sum(1).to(torch.int32)
        enc = m.encoder(ids, enc_valid)
        B = ids.shape[0]
        seq = torch.full((B, 1), cfg.decoder_start_token_id, dtype=torch.long)
        done = torch.zeros(B, dtype=torch.bool)
        for _ in range(max_length - 1):
            dec_valid = torch.full((B,), seq.shape[1], dtype=torch.int32)
            dec = m.decoder(seq, dec_valid, enc=enc, enc_valid=enc_valid)
            h = dec[:, -1] * (cfg.d_model ** -0.5)
            nxt = m.lm_head(h.float()).argmax(-1)
            nxt = torch.where(done, torch.full_like(nxt, cfg.pad_token_id), nxt)
            seq = torch.cat([seq, nxt.unsqueeze(1)], dim=1)
            done |= nxt == cfg.eos_token_id
            if bool(done.all()):
                break
        return seq

    @torch.no_grad()
    def test_greedy_matches_full_rerun(self):
        m, cfg = self._model()
        ids = torch.randint(3, cfg.vocab_size, (3, 24))
        ids[:, -1] = cfg.eos_token_id
        ids[0, 12:] = cfg.pad_token_id  # ragged source (cross-attn mask)
        got = m.generate(ids, max_length=12)
        want = self._reference_greedy(m, cfg, ids, 12)
        L = min(got.shape[1], want.shape[1])
        assert torch.equal(got[:, :L], want[:, :L]), (got, want)

    @torch.no_grad()
    def test_beam_runs_and_respects_eos(self):
        m, cfg = self._model()
        ids = torch.randint(3, cfg.vocab_size, (2, 16))
        ids[:, -1] = cfg.eos_token_id
        out = m.generate(ids, max_length=10, num_beams=3)
        assert out.shape[0] == 2 and out.shape[1] <= 10
