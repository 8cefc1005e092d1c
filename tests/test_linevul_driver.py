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
