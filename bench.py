#!/usr/bin/env python3
"""Flagship benchmark: DeepDFA flow-GNN training throughput (graphs/s).

Driver contract (see repo instructions): `python bench.py --gpus N --steps K
--warmup W`, launched under torch.distributed.run for N>1 (one rank per GPU
over RCCL). Measures the headline metric of BASELINE.json — training
throughput of the DeepDFA flow-GNN on Big-Vul-shaped synthetic CFG batches,
batch 256 graphs per GPU (weak scaling), bf16 compute + fp32 master Adam.
Baseline derived from the reference's published train time: ~810 graphs/s on
RTX 3090.
"""

import argparse
import json
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

from deepdfa_amd.data import parse_limits  # noqa: E402
from deepdfa_amd.graph import synthetic_cfg_batch  # noqa: E402
from deepdfa_amd.models import FlowGNNGGNNModule  # noqa: E402
from deepdfa_amd.parallel.ddp import DDPEngine, init_distributed, world_size  # noqa: E402

BASELINE_GRAPHS_PER_SEC = 810.0
FEAT = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000"


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=50)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--batch", type=int, default=256)
    ap.add_argument("--n-batches", type=int, default=4, help="pre-built batch pool size")
    ap.add_argument("--no-graph-capture", action="store_true")
    args = ap.parse_args()

    rank = init_distributed()
    ws = world_size()
    use_cuda = torch.cuda.is_available()
    device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0))) if use_cuda else torch.device("cpu")
    if use_cuda:
        torch.cuda.set_device(device)
    dtype = "bf16" if use_cuda else "fp32"

    torch.manual_seed(0)
    spec = parse_limits(FEAT)
    model = FlowGNNGGNNModule(
        feat=FEAT, input_dim=spec.input_dim, hidden_dim=32, n_steps=5, num_output_layers=3
    ).to(device)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3, weight_decay=1e-2)
    ddp = DDPEngine(model, bucket_cap_mb=64.0)

    # pre-build per-rank batch pool (different data per rank, fixed shapes per
    # batch: synthetic Big-Vul-shaped CFGs, ~45 nodes/graph avg)
    batches = [
        synthetic_cfg_batch(args.batch, seed=1000 * rank + i, input_dim=spec.input_dim).to(device)
        for i in range(args.n_batches)
    ]

    autocast = torch.autocast(device_type="cuda", dtype=torch.bfloat16) if use_cuda else None
    # hipGraph capture: the flow-GNN step is ~180 small kernels, so replaying
    # a captured graph removes the host launch overhead wholesale. One graph
    # per pre-built batch (shapes differ); eager path kept for multi-GPU.
    use_graphs = use_cuda and not args.no_graph_capture and ws == 1
    if use_graphs:
        opt = torch.optim.Adam(
            model.parameters(), lr=1e-3, weight_decay=1e-2, capturable=True, foreach=True
        )

    def step(i: int):
        g = batches[i % len(batches)]
        label = model.get_label(g)
        if autocast is not None:
            with autocast:
                logits = model(g, {})
        else:
            logits = model(g, {})
        loss = model.loss_fn(logits.float(), label)
        opt.zero_grad(set_to_none=not use_graphs)
        loss.backward()
        ddp.finalize()
        opt.step()
        return loss

    # warmup
    for i in range(args.warmup):
        step(i)

    graphs = []
    if use_graphs:
        torch.cuda.synchronize()
        pool = torch.cuda.graph_pool_handle()
        for i in range(len(batches)):
            cg = torch.cuda.CUDAGraph()
            with torch.cuda.graph(cg, pool=pool):
                step(i)
            graphs.append(cg)
        torch.cuda.synchronize()

        def step(i):  # noqa: F811 — replay path
            graphs[i % len(graphs)].replay()

    def barrier_sync():
        if ws > 1:
            torch.distributed.barrier()
        if use_cuda:
            torch.cuda.synchronize()

    barrier_sync()
    t0 = time.perf_counter()
    for i in range(args.steps):
        step(i)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], device=device if use_cuda else "cpu")
    if ws > 1:
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed = float(t.item())

    total_graphs = args.batch * args.steps * ws
    gps = total_graphs / elapsed
    if rank == 0:
        print(
            json.dumps(
                {
                    "metric": "train_graphs_per_sec",
                    "value": gps,
                    "unit": "graphs/s",
                    "n_gpus": ws,
                    "steps": args.steps,
                    "warmup": args.warmup,
                    "ms_per_step": elapsed / args.steps * 1000.0,
                    "higher_is_better": True,
                    "scaling": "weak",
                    "vs_baseline": gps / BASELINE_GRAPHS_PER_SEC,
                    "dtype": dtype,
                    "data": "synthetic",
                    "config": {
                        "model": "DeepDFA-FlowGNN-GGNN(n_steps=5,D=128)",
                        "global_batch": args.batch * ws,
                        "seq_len": None,
                        "avg_nodes_per_graph": 45,
                        "parallelism": f"dp{ws}",
                    },
                }
            )
        )
    if ws > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
