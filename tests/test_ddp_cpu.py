"""DDP engine correctness on CPU: 2 processes over gloo.

Verifies the RCCL-path semantics with the gloo backend (world_size 2, file
rendezvous): initial parameter broadcast, bucketed gradient all-reduce
producing the exact average of per-rank gradients, rank-identical
parameters after an optimizer step.
"""

import os
import subprocess
import sys

import torch

from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
from deepdfa_amd.models import FlowGNNGGNNModule

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _single_process_grads(seed_data):
    torch.manual_seed(100)  # rank0 init (the broadcast source)
    model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=32, n_steps=2, num_output_layers=3)
    g = synthetic_cfg_batch(8, seed=seed_data)
    loss = model.training_step((g, {}))
    loss.backward()
    return torch.cat([p.grad.flatten() for p in model.parameters() if p.grad is not None])


def _run_workers(tmp_path, mode):
    f = str(tmp_path / "rendezvous")
    outs_paths = [str(tmp_path / f"out{r}.pt") for r in range(2)]
    procs = [
        subprocess.Popen(
            [sys.executable, os.path.join(REPO, "tests", "ddp_worker.py"),
             str(r), "2", f, outs_paths[r], mode],
            env={**os.environ, "PYTHONPATH": REPO},
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
        )
        for r in range(2)
    ]
    for p in procs:
        out, _ = p.communicate(timeout=240)
        assert p.returncode == 0, out.decode()[-2000:]
    return [torch.load(p, weights_only=True) for p in outs_paths]


def test_ddp_two_ranks(tmp_path):
    outs = _run_workers(tmp_path, "basic")
    # 1. both ranks ended with identical params (broadcast + identical update)
    assert torch.allclose(outs[0]["params"], outs[1]["params"], atol=1e-7)
    # 2. grads identical across ranks and equal to the average of per-rank grads
    assert torch.allclose(outs[0]["grad"], outs[1]["grad"], atol=1e-7)
    expected = (_single_process_grads(0) + _single_process_grads(1)) / 2
    assert torch.allclose(outs[0]["grad"], expected, atol=1e-5)


def test_ddp_gradient_accumulation(tmp_path):
    """no_sync() micro-batches accumulate locally; the boundary reduce must
    average the SUM of all micro-batch grads (advisor round-1 high finding:
    the hook engine used to reduce after micro-batch 1 and discard the rest)."""
    outs = _run_workers(tmp_path, "accum")
    assert torch.allclose(outs[0]["grad"], outs[1]["grad"], atol=1e-7)
    expected = sum(
        _single_process_grads(r * 10 + m) for r in range(2) for m in range(3)
    ) / 2
    assert torch.allclose(outs[0]["grad"], expected, atol=1e-5)


def test_ddp_skip_participation(tmp_path):
    """A rank whose batch was entirely missing still participates in the
    collectives via finalize() (zero contribution) — no hang, and the
    average is rank0's grads / world_size."""
    outs = _run_workers(tmp_path, "skip")
    assert torch.allclose(outs[0]["grad"], outs[1]["grad"], atol=1e-7)
    expected = _single_process_grads(0) / 2
    assert torch.allclose(outs[0]["grad"], expected, atol=1e-5)
    assert torch.allclose(outs[0]["params"], outs[1]["params"], atol=1e-7)
