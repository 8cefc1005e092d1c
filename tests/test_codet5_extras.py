"""CodeT5 extras: clone model, generation + beam, CodeBLEU, run_exp matrix."""

import torch

from deepdfa_amd.evaluator import bleu, calc_code_bleu, smoothed_bleu4
from deepdfa_amd.models.codet5 import CloneModel
from deepdfa_amd.models.t5 import T5Config
from deepdfa_amd.train import run_clone, run_gen


def small_cfg():
    return T5Config(vocab_size=300, d_model=64, d_kv=16, d_ff=128,
                    num_layers=1, num_decoder_layers=1, num_heads=4)


def test_clone_model_forward():
    torch.manual_seed(0)
    model = CloneModel(config=small_cfg(), max_source_length=20)
    ids = torch.randint(3, 300, (2, 40))
    ids[:, 18] = 2
    ids[:, 19] = 0
    ids[:, 38] = 2
    ids[:, 39] = 0
    loss, prob = model(ids, labels=torch.tensor([0, 1]))
    assert prob.shape == (2, 2)
    loss.backward()


def test_generate_greedy_and_beam():
    torch.manual_seed(0)
    from deepdfa_amd.models.t5 import T5ForConditionalGeneration

    m = T5ForConditionalGeneration(small_cfg()).eval()
    ids = torch.randint(3, 300, (2, 12))
    out_g = m.generate(ids, max_length=6, num_beams=1)
    assert out_g.shape[0] == 2 and out_g.shape[1] <= 6
    out_b = m.generate(ids, max_length=6, num_beams=3)
    assert out_b.shape[0] == 2


def test_bleu_properties():
    ref = "int main ( ) { return 0 ; }".split()
    assert smoothed_bleu4(ref, ref) > 0.95
    assert smoothed_bleu4(ref, "totally different words here".split()) < 0.2
    assert bleu([ref], [ref]) > 0.95
    r = calc_code_bleu(["int x = 1 ;"], ["int x = 1 ;"])
    assert r["code_bleu"] > 0.9
    assert r["components_used"] == 2  # tree_sitter gated off here
    # keyword weighting: a keyword MATCH outweighs an identifier match at
    # the same number of token mismatches
    kw_match = calc_code_bleu(["return x ;"], ["return y ;"])
    kw_miss = calc_code_bleu(["return x ;"], ["goto x ;"])
    assert kw_match["weighted_ngram_match"] > kw_miss["weighted_ngram_match"]


def test_run_gen_end_to_end(tmp_path):
    res = run_gen.main([
        "--do_train", "--do_test", "--num_train_epochs", "1",
        "--n_synthetic", "16", "--num_layers", "1", "--d_model", "64",
        "--max_source_length", "48", "--max_target_length", "12",
        "--beam_size", "2", "--output_dir", str(tmp_path / "gen"),
    ])
    assert "train_loss" in res and "bleu4" in res
    assert 0.0 <= res["bleu4"] <= 1.0


def test_run_clone_end_to_end(tmp_path):
    res = run_clone.main([
        "--do_train", "--do_test", "--num_train_epochs", "1",
        "--n_synthetic", "16", "--num_layers", "1", "--d_model", "64",
        "--max_source_length", "32", "--output_dir", str(tmp_path / "clone"),
    ])
    assert "train_loss" in res and "test_acc" in res
