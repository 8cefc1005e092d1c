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


def test_ddp_two_ranks(tmp_path):
    f = str(tmp_path / "rendezvous")
    outs_paths = [str(tmp_path / f"out{r}.pt") for r in range(2)]
    procs = [
        subprocess.Popen(
            [sys.executable, os.path.join(REPO, "tests", "ddp_worker.py"), str(r), "2", f, outs_paths[r]],
            env={**os.environ, "PYTHONPATH": REPO},
            stdout=subprocess.PIPE,
            stderr=subprocess.STDOUT,
        )
        for r in range(2)
    ]
    logs = []
    for p in procs:
        out, _ = p.communicate(timeout=240)
        logs.append(out.decode())
        assert p.returncode == 0, out.decode()[-2000:]
    outs = [torch.load(p, weights_only=True) for p in outs_paths]
    # 1. both ranks ended with identical params (broadcast + identical update)
    assert torch.allclose(outs[0]["params"], outs[1]["params"], atol=1e-7)
    # 2. grads identical across ranks and equal to the average of per-rank grads
    assert torch.allclose(outs[0]["grad"], outs[1]["grad"], atol=1e-7)
    expected = (_single_process_grads(0) + _single_process_grads(1)) / 2
    assert torch.allclose(outs[0]["grad"], expected, atol=1e-5)
