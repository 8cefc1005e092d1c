"""End-to-end CodeT5 run_defect driver test (CPU, tiny config)."""

import os

from deepdfa_amd.train import run_defect


def test_run_defect_train_test(tmp_path):
    out = str(tmp_path / "defect")
    results = run_defect.main(
        [
            "--do_train",
            "--do_test",
            "--flowgnn_model",
            "--flowgnn_data",
            "--num_train_epochs",
            "1",
            "--num_layers",
            "1",
            "--d_model",
            "64",
            "--max_source_length",
            "64",
            "--train_batch_size",
            "8",
            "--eval_batch_size",
            "8",
            "--gradient_accumulation_steps",
            "2",
            "--n_synthetic",
            "80",
            "--output_dir",
            out,
        ]
    )
    assert os.path.exists(os.path.join(out, "checkpoint-best-acc", "pytorch_model.bin"))
    assert os.path.exists(os.path.join(out, "checkpoint-last", "pytorch_model.bin"))
    assert "test" in results and "eval_f1" in results["test"]
