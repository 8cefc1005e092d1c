#!/usr/bin/env python3
"""Experiment runner (reference CodeT5/sh/run_exp.py:21-108 parity):
per-task hyperparameter matrix dispatching the task drivers.

  python scripts/run_exp.py --model_tag codet5_base --task defect [--flowgnn_model]
"""

import argparse
import subprocess
import sys

TASK_PARAMS = {
    # task: (max_source, max_target, lr, epochs, batch, accum, patience)
    "defect": dict(src_len=512, trg_len=3, lr=2e-5, epochs=10, bs=8, accum=4, patience=2),
    "clone": dict(src_len=400, trg_len=400, lr=5e-5, epochs=1, bs=8, accum=1, patience=2),
    "summarize": dict(src_len=256, trg_len=128, lr=5e-5, epochs=15, bs=32, accum=1, patience=2),
    "translate": dict(src_len=320, trg_len=256, lr=5e-5, epochs=100, bs=16, accum=1, patience=5),
    "refine": dict(src_len=130, trg_len=120, lr=5e-5, epochs=50, bs=16, accum=1, patience=5),
    "concode": dict(src_len=320, trg_len=150, lr=5e-5, epochs=30, bs=16, accum=1, patience=3),
}


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model_tag", default="codet5_base")
    p.add_argument("--task", required=True, choices=sorted(TASK_PARAMS))
    p.add_argument("--flowgnn_data", action="store_true")
    p.add_argument("--flowgnn_model", action="store_true")
    p.add_argument("extra", nargs="*")
    args = p.parse_args()
    hp = TASK_PARAMS[args.task]
    if args.task == "defect":
        cmd = [sys.executable, "-m", "deepdfa_amd.train.run_defect", "--do_train",
               "--do_eval", "--do_test",
               "--max_source_length", str(hp["src_len"]),
               "--learning_rate", str(hp["lr"]),
               "--num_train_epochs", str(hp["epochs"]),
               "--train_batch_size", str(hp["bs"]),
               "--gradient_accumulation_steps", str(hp["accum"]),
               "--patience", str(hp["patience"])]
        if args.flowgnn_data:
            cmd.append("--flowgnn_data")
        if args.flowgnn_model:
            cmd += ["--flowgnn_model", "--flowgnn_data"]
    elif args.task == "clone":
        cmd = [sys.executable, "-m", "deepdfa_amd.train.run_clone", "--do_train",
               "--do_test", "--learning_rate", str(hp["lr"]),
               "--num_train_epochs", str(hp["epochs"]),
               "--train_batch_size", str(hp["bs"])]
    else:
        cmd = [sys.executable, "-m", "deepdfa_amd.train.run_gen", "--do_train",
               "--do_test", "--task", args.task,
               "--max_source_length", str(hp["src_len"]),
               "--max_target_length", str(hp["trg_len"]),
               "--learning_rate", str(hp["lr"]),
               "--num_train_epochs", str(hp["epochs"]),
               "--train_batch_size", str(hp["bs"])]
    cmd += args.extra
    print("run_exp:", " ".join(cmd))
    raise SystemExit(subprocess.call(cmd))


if __name__ == "__main__":
    main()
