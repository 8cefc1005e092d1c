"""Standalone DDP worker for tests/test_ddp_cpu.py (gloo, file rendezvous)."""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from deepdfa_amd.graph.synthetic import synthetic_cfg_batch  # noqa: E402
from deepdfa_amd.models import FlowGNNGGNNModule  # noqa: E402


def main():
    rank, world = int(sys.argv[1]), int(sys.argv[2])
    file_name, out_path = sys.argv[3], sys.argv[4]
    mode = sys.argv[5] if len(sys.argv) > 5 else "basic"
    dist.init_process_group("gloo", init_method=f"file://{file_name}", rank=rank, world_size=world)
    os.environ["WORLD_SIZE"] = str(world)
    from deepdfa_amd.parallel.ddp import DDPEngine

    torch.manual_seed(100 + rank)  # deliberately different init per rank
    model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=32, n_steps=2, num_output_layers=3)
    ddp = DDPEngine(model, bucket_cap_mb=0.25)  # force multiple buckets
    assert ddp.enabled and len(ddp.buckets) > 1
    if mode == "accum":
        # gradient accumulation: 2 no_sync micro-batches + 1 sync micro-batch;
        # reduced grad must be mean over ranks of the SUM over micro-batches
        for micro in range(3):
            g = synthetic_cfg_batch(8, seed=rank * 10 + micro)
            loss = model.training_step((g, {}))
            if micro < 2:
                with ddp.no_sync():
                    loss.backward()
            else:
                loss.backward()
        ddp.finalize()
    elif mode == "skip":
        # rank 1 skips its backward entirely (all-graphs-missing batch) but
        # still participates in the collectives via finalize()
        if rank == 0:
            g = synthetic_cfg_batch(8, seed=0)
            loss = model.training_step((g, {}))
            loss.backward()
        ddp.finalize()
    else:
        g = synthetic_cfg_batch(8, seed=rank)  # different data per rank
        loss = model.training_step((g, {}))
        loss.backward()
        ddp.finalize()
    grad_vec = torch.cat([p.grad.flatten() for p in model.parameters() if p.grad is not None])
    opt = torch.optim.SGD(model.parameters(), lr=0.1)
    opt.step()
    pvec = torch.cat([p.detach().flatten() for p in model.parameters()])
    torch.save({"grad": grad_vec, "params": pvec}, out_path)
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
