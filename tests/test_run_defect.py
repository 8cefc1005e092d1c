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


def test_run_defect_early_stop(tmp_path, caplog):
    """Patience-based early stopping (reference run_defect.py:398-405):
    with patience 1 and enough epochs, the loop must stop before
    num_train_epochs once eval F1 stops improving."""
    import logging

    out = str(tmp_path / "defect_es")
    with caplog.at_level(logging.INFO):
        run_defect.main(
            [
                "--do_train", "--num_train_epochs", "6", "--num_layers", "1",
                "--d_model", "64", "--max_source_length", "64",
                "--train_batch_size", "8", "--eval_batch_size", "8",
                "--n_synthetic", "60", "--patience", "1",
                "--output_dir", out,
            ]
        )
    epochs_run = sum(1 for r in caplog.records if r.getMessage().startswith("epoch "))
    stopped = any("early stop" in r.getMessage() for r in caplog.records)
    assert stopped or epochs_run == 6  # stops early unless F1 kept improving
    if stopped:
        assert epochs_run < 6
