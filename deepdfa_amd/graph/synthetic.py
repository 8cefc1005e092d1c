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
    # vectorized edge draw (the per-node scalar RNG loop cost ~1 ms/graph
    # and dominated the CPU dataloader epoch — ~50 scalar torch RNG calls
    # at ~10 us each): same topology distribution, one RNG call per family
    i = torch.arange(1, n)
    cap = torch.clamp(i, max=3)
    back = 1 + (torch.rand(n - 1, generator=gen) * cap).to(torch.int64)
    src_t = [i - back]
    dst_t = [i]
    r = torch.rand(n - 1, generator=gen)
    fwd = (r < 0.15) & (i + 2 < n)  # forward branch
    skip = torch.randint(2, 6, (n - 1,), generator=gen)
    if bool(fwd.any()):
        src_t.append(i[fwd])
        dst_t.append(torch.clamp(i[fwd] + skip[fwd], max=n - 1))
    loop = (r >= 0.15) & (r < 0.22) & (i > 4)  # back edge (loop)
    u2 = torch.rand(n - 1, generator=gen)
    if bool(loop.any()):
        src_t.append(i[loop])
        dst_t.append((u2[loop] * (i[loop] - 1).float()).to(torch.int64))
    src: List[int] = torch.cat(src_t).tolist()
    dst: List[int] = torch.cat(dst_t).tolist()

    ndata = {}
    # ~60% of statements are not definitions -> feature index 0; others draw
    # from a skewed distribution over [1, input_dim). All four subkey
    # features in ONE (n, 4) draw (scalar-call overhead dominated).
    import math

    is_def = torch.rand(n, 4, generator=gen) < 0.4
    u = torch.rand(n, 4, generator=gen)
    vocab = torch.clamp(
        torch.exp(u * math.log(input_dim - 1)).to(torch.int64), 1, input_dim - 1
    )
    feats = torch.where(is_def, vocab, torch.zeros((), dtype=torch.int64))
    for j, feat in enumerate(ALL_FEATS):
        ndata[f"_ABS_DATAFLOW_{feat}"] = feats[:, j].contiguous()
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
