"""Seq2Seq (CodeBERT-style enc-dec) + run_multi_gen driver."""

import numpy as np
import torch

from deepdfa_amd.models.roberta import RobertaConfig
from deepdfa_amd.models.seq2seq import Seq2Seq
from deepdfa_amd.train import run_multi_gen


def tiny_cfg():
    return RobertaConfig(
        vocab_size=200, hidden_size=32, num_hidden_layers=1,
        num_attention_heads=4, intermediate_size=64, max_position_embeddings=96,
    )


def make_model(**kw):
    torch.manual_seed(0)
    return Seq2Seq(tiny_cfg(), num_decoder_layers=2, beam_size=3,
                   max_length=8, sos_id=0, eos_id=2, **kw)


def test_seq2seq_train_loss_and_grads():
    m = make_model()
    src = torch.randint(3, 200, (2, 12))
    tgt = torch.randint(3, 200, (2, 6))
    tgt[:, 0] = 0  # sos
    tgt[:, -1] = 1  # padding (RoBERTa pad_token_id=1)
    loss, loss_sum, n_active = m(src, target_ids=tgt)
    assert loss.requires_grad and float(loss.detach()) > 0
    assert int(n_active) == int(tgt.ne(1)[..., 1:].sum())
    loss.backward()
    assert m.decoder[0].self_attn.query.weight.grad is not None
    assert m.encoder.embeddings.word_embeddings.weight.grad is not None


def test_seq2seq_tied_lm_head():
    m = make_model()
    assert m.lm_head.weight.data_ptr() == m.encoder.embeddings.word_embeddings.weight.data_ptr()


def test_seq2seq_causal_decoder():
    """Changing a later target token must not change earlier logits."""
    m = make_model().eval()
    src = torch.randint(3, 200, (1, 12))
    tgt = torch.randint(3, 200, (1, 6))
    with torch.no_grad():
        mem, _ = m.encoder(src)
        mv = src.ne(0).sum(dim=1).to(torch.int32)
        a = m._decode(tgt, mem, mv)
        tgt2 = tgt.clone()
        tgt2[0, 4] = (tgt2[0, 4] + 7) % 200
        b = m._decode(tgt2, mem, mv)
    assert torch.allclose(a[0, :4], b[0, :4], atol=1e-5)
    assert not torch.allclose(a[0, 4:], b[0, 4:], atol=1e-5)


def test_seq2seq_beam_decode_shape():
    m = make_model().eval()
    src = torch.randint(3, 200, (2, 12))
    preds = m(src)
    assert preds.shape == (2, 3, m.max_length)
    assert (preds >= 0).all()
    # nothing after the first eos (zeros only)
    for b in range(2):
        row = preds[b, 0].tolist()
        if 2 in row:
            i = row.index(2)
            assert all(x == 0 for x in row[i:])


def test_sampling_probs_temperature():
    p = run_multi_gen.sampling_probs([100, 10])
    assert abs(p.sum() - 1) < 1e-9
    # temperature 0.7 flattens vs proportional
    assert p[1] > 10 / 110
    assert p[0] > p[1]


def test_run_multi_gen_end_to_end(tmp_path):
    res = run_multi_gen.main([
        "--tasks", "summarize,translate", "--max_steps", "6", "--eval_every", "3",
        "--n_synthetic", "6", "--train_batch_size", "4", "--num_layers", "1",
        "--d_model", "64", "--max_source_length", "32", "--max_target_length", "12",
        "--output_dir", str(tmp_path / "mg"),
    ])
    assert res["steps"] == 6
    assert set(res["best_bleu"]) == {"summarize", "translate"}
    assert all(v is None or np.isfinite(v) for v in res["train_loss"].values())
    assert (tmp_path / "mg" / "checkpoint-last.bin").exists()
