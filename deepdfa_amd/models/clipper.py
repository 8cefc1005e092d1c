"""Differentiable bitwise-union "meet" operators.

Parity target: reference DDFA/code_gnn/models/clipper.py:1-158 — the
earlier DeepDFA model iterations represented reaching-definition sets as
{0,1}-vectors and needed a differentiable set-union for the CFG meet
operator. Kept for capability parity (the GGNN path does not use them),
with the same numerical properties the reference's embedded pytest
functions assert: union(a, b) == min(a + b, 1) elementwise on {0,1}
inputs, gradients defined everywhere.
"""

from __future__ import annotations

import torch


def simple_union(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """a + b - a*b: exact on {0,1}, smooth in between (clipper.py:6-14)."""
    return a + b - a * b


def relu_union(a: torch.Tensor, b: torch.Tensor) -> torch.Tensor:
    """1 - relu(1 - a - b): exact on {0,1}, piecewise-linear
    (clipper.py:17-26)."""
    return 1.0 - torch.relu(1.0 - a - b)


def union_reduce(xs: torch.Tensor, dim: int = 0, kind: str = "simple") -> torch.Tensor:
    """Fold a union over a dimension (the DGL mailbox-reduce factory,
    clipper.py:53-90 equivalent)."""
    fn = simple_union if kind == "simple" else relu_union
    out = None
    for i in range(xs.shape[dim]):
        cur = xs.select(dim, i)
        out = cur if out is None else fn(out, cur)
    return out
