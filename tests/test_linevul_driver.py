"""End-to-end LineVul driver test (CPU, small config): reference CLI
surface, checkpoint layout, flow-GNN join, profiling jsonl contract."""

import json
import os

import torch

from deepdfa_amd.train import linevul_main


def test_linevul_main_train_test(tmp_path):
    out = str(tmp_path / "saved_models")
    results = linevul_main.main(
        [
            "--do_train",
            "--do_test",
            "--epochs",
            "1",
            "--num_layers",
            "2",
            "--block_size",
            "128",
            "--train_batch_size",
            "8",
            "--eval_batch_size",
            "1",
            "--n_synthetic",
            "120",
            "--seed",
            "1",
            "--profile",
            "--time",
            "--output_dir",
            out,
        ]
    )
    # reference checkpoint layout: checkpoint-best-f1/<seed>_linevul.bin
    ckpt = os.path.join(out, "checkpoint-best-f1", "1_linevul.bin")
    assert os.path.exists(ckpt)
    assert os.path.exists(os.path.join(out, "checkpoint-last", "model.bin"))
    sd = torch.load(ckpt, weights_only=True)
    assert any(k.startswith("encoder.") for k in sd)
    assert any(k.startswith("flowgnn_encoder.") for k in sd)
    assert "test" in results and "accuracy" in results["test"]
    # profiling jsonl contract
    prof_lines = open(os.path.join(out, "profiledata.jsonl")).read().strip().splitlines()
    assert prof_lines
    row = json.loads(prof_lines[0])
    assert {"flops", "macs", "params", "batch_size"} <= set(row)
    assert os.path.exists(os.path.join(out, "timedata.jsonl"))


def test_linevul_main_no_flowgnn(tmp_path):
    out = str(tmp_path / "nofg")
    linevul_main.main(
        [
            "--do_train",
            "--epochs",
            "1",
            "--num_layers",
            "1",
            "--block_size",
            "128",
            "--train_batch_size",
            "8",
            "--n_synthetic",
            "40",
            "--no_flowgnn",
            "--output_dir",
            out,
        ]
    )
    assert os.path.exists(os.path.join(out, "checkpoint-last", "model.bin"))


def test_end_to_end_learning_reaches_high_f1(tmp_path):
    """The synthetic vulnerability signal (strcpy pattern) must be learned
    end-to-end through the real driver: best eval F1 > 0.8 in a few
    epochs on a small encoder (seed-fixed, CPU)."""
    import torch

    from deepdfa_amd.data.text_dataset import TextDataset
    from deepdfa_amd.data.tokenization import HashTokenizer
    from deepdfa_amd.models.linevul import Model
    from deepdfa_amd.models.roberta import RobertaConfig
    from deepdfa_amd.train import linevul_main

    args = linevul_main.build_args([
        "--do_train", "--no_flowgnn", "--epochs", "6", "--n_synthetic", "400",
        "--num_layers", "2", "--block_size", "64", "--train_batch_size", "16",
        "--learning_rate", "1e-4", "--output_dir", str(tmp_path),
    ])
    torch.manual_seed(1)
    cfg = RobertaConfig(num_hidden_layers=2)
    tok = HashTokenizer(vocab_size=cfg.vocab_size)
    model = Model(config=cfg)
    dss = {p: TextDataset(tok, args, partition=p, block_size=64, n_synthetic=400)
           for p in ("train", "val")}
    best = linevul_main.train(args, model, dss["train"], dss["val"], None,
                              torch.device("cpu"))
    assert best > 0.8, best


def test_shipped_tokenizer_assets_load():
    """The checked-in bpe/word-level tokenizer JSONs (assets/, reference
    LineVul ships the same two) load and round-trip C-ish source."""
    import os

    from deepdfa_amd.train.linevul_main import build_args, build_tokenizer
    from deepdfa_amd.models.roberta import RobertaConfig

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    for flag, kind in (("--use_non_pretrained_tokenizer", "bpe_tokenizer"),
                       ("--use_word_level_tokenizer", "word_level_tokenizer")):
        assert os.path.exists(os.path.join(repo, "assets", kind, "tokenizer.json"))
        args = build_args([flag])
        cfg = RobertaConfig(vocab_size=50265)
        tok = build_tokenizer(args, cfg)
        ids = tok.encode("int main(void) { return strcpy(dst, src); }", 64)
        assert len(ids) == 64 and max(ids) < 50265
