"""Synthetic Big-Vul-shaped CFG generator.

There is no network access for the real Big-Vul dataset, so benchmarks and
tests run on synthetic CFGs shaped like the reference's preprocessed data
(BASELINE.json: "synthetic Big-Vul-shaped CFGs / random-init weights"):

  * node count per function: log-normal, clamped to [3, 500], mean ~45
    (the reference dataset averages ~40-50 CFG nodes per function; range
    ~5-500 — SURVEY.md §7 "Hard parts" item 2);
  * edges: CFGs are mostly sequential with branches — each node i>0 gets an
    in-edge from a recent predecessor, plus a back-edge (loop) or forward
    branch with small probability; self-loops added on every node exactly as
    the reference's dbize_graphs.py:25 does;
  * node features: 4 abstract-dataflow subkey indices in [0, input_dim):
    index 0 = "not a definition" (majority of nodes), 1 = UNKNOWN, the rest
    a Zipf-ish draw over the vocabulary (mirrors dbize_absdf.py:35-42
    semantics where most statements are not assignments);
  * _VULN: per-node binary label, sparse; graph label = max over nodes
    (reference base_module.py:87-88).
"""

from __future__ import annotations

from typing import List, Optional

import torch

from .batch import BatchedCFG, batch_graphs

ALL_FEATS = ["api", "datatype", "literal", "operator"]


def synthetic_cfg(
    gen: torch.Generator,
    input_dim: int = 1002,
    mean_nodes: float = 45.0,
    vuln_rate: float = 0.06,
) -> BatchedCFG:
    """One synthetic CFG with reference-shaped ndata."""
    n = int(
        torch.clamp(
            torch.exp(torch.normal(torch.tensor(3.55), torch.tensor(0.75), generator=gen)),
            3,
            500,
        ).item()
    )
    src: List[int] = []
    dst: List[int] = []
    for i in range(1, n):
        # sequential edge from a recent predecessor (branch join pattern)
        back = int(torch.randint(1, min(i, 3) + 1, (1,), generator=gen).item())
        src.append(i - back)
        dst.append(i)
        r = torch.rand((), generator=gen).item()
        if r < 0.15 and i + 2 < n:  # forward branch
            skip = int(torch.randint(2, 6, (1,), generator=gen).item())
            src.append(i)
            dst.append(min(i + skip, n - 1))
        elif r < 0.22 and i > 4:  # back edge (loop)
            tgt = int(torch.randint(0, i - 1, (1,), generator=gen).item())
            src.append(i)
            dst.append(tgt)

    ndata = {}
    # ~60% of statements are not definitions -> feature index 0; others draw
    # from a skewed distribution over [1, input_dim).
    for feat in ALL_FEATS:
        is_def = torch.rand(n, generator=gen) < 0.4
        # Zipf-ish: floor(exp(U*log(input_dim-1)))
        u = torch.rand(n, generator=gen)
        vocab = torch.clamp(
            torch.exp(u * torch.log(torch.tensor(float(input_dim - 1)))).to(torch.int64),
            1,
            input_dim - 1,
        )
        ndata[f"_ABS_DATAFLOW_{feat}"] = torch.where(
            is_def, vocab, torch.zeros(n, dtype=torch.int64)
        )
    ndata["_VULN"] = (torch.rand(n, generator=gen) < vuln_rate).to(torch.int64)
    return BatchedCFG.from_edges(n, src, dst, ndata=ndata, add_self_loops=True)


def synthetic_cfg_batch(
    batch_size: int,
    seed: int = 0,
    input_dim: int = 1002,
    mean_nodes: float = 45.0,
    device: Optional[torch.device] = None,
) -> BatchedCFG:
    gen = torch.Generator().manual_seed(seed)
    g = batch_graphs(
        [synthetic_cfg(gen, input_dim=input_dim, mean_nodes=mean_nodes) for _ in range(batch_size)]
    )
    if device is not None:
        g = g.to(device)
    return g
