"""Driver-contract regression tests for bench.py: single-process CPU run
and a 2-process gloo torchrun launch must both print the JSON line."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _check_json(line: str, n_gpus: int):
    row = json.loads(line)
    assert row["metric"] == "train_graphs_per_sec"
    assert row["n_gpus"] == n_gpus
    assert row["value"] > 0 and row["ms_per_step"] > 0
    assert row["scaling"] == "weak"
    assert row["config"]["global_batch"] == 8 * n_gpus
    return row


def test_bench_single_process_cpu():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--model", "ddfa", "--batch", "8", "--n-batches", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    _check_json(out.stdout.strip().splitlines()[-1], 1)


def test_bench_torchrun_two_ranks_gloo():
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29619", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--model", "ddfa", "--batch", "8", "--n-batches", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1  # rank 0 only
    _check_json(json_lines[-1], 2)
