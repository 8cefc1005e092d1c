"""Standalone 2-rank Trainer.fit worker (gloo) for tests/test_trainer_ddp.py."""

import os
import sys

import torch
import torch.distributed as dist

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from deepdfa_amd.data.datamodule import BigVulDatasetLineVDDataModule  # noqa: E402
from deepdfa_amd.models import FlowGNNGGNNModule  # noqa: E402
from deepdfa_amd.train.trainer import Trainer  # noqa: E402


def main():
    rank, world = int(sys.argv[1]), int(sys.argv[2])
    file_name, out_path, root = sys.argv[3], sys.argv[4], sys.argv[5]
    dist.init_process_group("gloo", init_method=f"file://{file_name}", rank=rank, world_size=world)
    os.environ["WORLD_SIZE"] = str(world)
    torch.manual_seed(7 + rank)  # different init per rank; DDP must broadcast
    dm = BigVulDatasetLineVDDataModule(
        batch_size=16, n_synthetic=300, undersample="v1.0", seed=0
    )
    model = FlowGNNGGNNModule(
        input_dim=dm.input_dim, hidden_dim=8, n_steps=2, num_output_layers=2
    )
    trainer = Trainer(
        max_epochs=2, default_root_dir=os.path.join(root, f"rank{rank}" if rank else "run"),
        precision="fp32", seed=0, periodic_every=100, graph_capture=False,
    )
    # rank0's dir holds the real checkpoints; other ranks write nothing
    trainer.ckpt_dir = os.path.join(root, "ckpts")
    os.makedirs(trainer.ckpt_dir, exist_ok=True)
    out = trainer.fit(model, dm)
    torch.save(
        {
            "params": torch.cat([p.detach().flatten() for p in model.parameters()]),
            "history": out["history"],
            "best": out["best_checkpoint"],
        },
        out_path,
    )
    dist.barrier()
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
