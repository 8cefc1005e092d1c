"""Run only wgrad2 + gemm2 hot shapes for a PMC capture (see flash_pmc_probe)."""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepdfa_amd.ops import load_ext

ext = load_ext(required=True)
dev = "cuda"
bf = torch.bfloat16
torch.manual_seed(0)
N, D, FF = 8192, 768, 3072
x = torch.randn(N, D, device=dev, dtype=bf) * 0.3
w = torch.randn(D, D, device=dev, dtype=bf) * 0.05
wff = torch.randn(FF, D, device=dev, dtype=bf) * 0.05
dy = torch.randn(N, D, device=dev, dtype=bf) * 0.3
dyff = torch.randn(N, FF, device=dev, dtype=bf) * 0.3
b32 = torch.randn(D, device=dev)
torch.cuda.synchronize()
for _ in range(30):
    ext.wgrad(dy, x)
    ext.wgrad(dyff, x)
    ext.gemm2(x, w, b32, None)
    ext.gemm2(x, wff, None, None)
torch.cuda.synchronize()
print("done")
