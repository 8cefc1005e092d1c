"""Within-probe interleaved kernel microbenchmark (guide §5.4 rule 24).

Times each hot kernel of the LineVul step at its exact shapes, interleaved
rounds in one process, medians reported — the definitive per-kernel numbers
on one box/clock. Run on a GPU box: python tools/kernel_probe.py
"""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepdfa_amd.ops import load_ext

assert torch.cuda.is_available()
ext = load_ext(required=True)
dev = "cuda"
bf = torch.bfloat16
torch.manual_seed(0)

N, D, FF, L, B, H = 8192, 768, 3072, 512, 16, 12

x = torch.randn(N, D, device=dev, dtype=bf) * 0.3
w = torch.randn(D, D, device=dev, dtype=bf) * 0.05
wff = torch.randn(FF, D, device=dev, dtype=bf) * 0.05
dy = torch.randn(N, D, device=dev, dtype=bf) * 0.3
dyff = torch.randn(N, FF, device=dev, dtype=bf) * 0.3
b32 = torch.randn(D, device=dev)
bff = torch.randn(FF, device=dev)
q = torch.randn(B, L, H * 64, device=dev, dtype=bf) * 0.3
k = torch.randn(B, L, H * 64, device=dev, dtype=bf) * 0.3
v = torch.randn(B, L, H * 64, device=dev, dtype=bf) * 0.3
dO = torch.randn(B, L, H * 64, device=dev, dtype=bf) * 0.3
valid = torch.full((B,), L, dtype=torch.int32, device=dev)
O, lse = ext.flash_attn_fwd(q, k, v, H, valid, None, 0.125, False, 0.0, 0)
Dt = torch.empty(B, H, L, device=dev)
xff = torch.randn(N, FF, device=dev, dtype=bf)
_, lnm, lnr = ext.layernorm_fwd(x, b32, b32, 1e-5)

wf32 = w.float()
xf32 = x.float()
dyf32 = dy.float()

variants = {
    "wgrad2 (768,768,K8192)": lambda: ext.wgrad(dy, x),
    "wgrad2 (3072,768,K8192)": lambda: ext.wgrad(dyff, x),
    "matmul wgrad bf16 (hipblaslt)": lambda: torch.matmul(dy.t(), x),
    "gemm2 fwd (8192x768x768)": lambda: ext.gemm2(x, w, b32, None),
    "gemm2 ffn (8192x3072 K768)": lambda: ext.gemm2(x, wff, bff, None),
    "gemm2 wo (8192x768 K3072)": lambda: ext.gemm2(xff, wff.t().contiguous(), b32, None),
    "matmul fwd bf16 (hipblaslt)": lambda: torch.matmul(x, w.t()),
    "flash fwd": lambda: ext.flash_attn_fwd(q, k, v, H, valid, None, 0.125, False, 0.0, 0),
    "flash bwd (dterm+dq+dkv)": lambda: ext.flash_attn_bwd(
        dO, q, k, v, O, lse, H, valid, None, 0.125, False, 0.0, 0, False, False),
    "bias_gelu fwd (8192x3072)": lambda: ext.bias_gelu_fwd(xff, bff),
    "bias_gelu bwd": lambda: ext.bias_gelu_bwd(dyff, xff, bff),
    "colsum (8192x768)": lambda: ext.colsum(dy),
    "layernorm fwd (8192x768)": lambda: ext.layernorm_fwd(x, b32, b32, 1e-5),
    "layernorm bwd": lambda: ext.layernorm_bwd(dy, x, b32, lnm, lnr),
    "layernorm wgrad": lambda: ext.layernorm_wgrad(dy, x, lnm, lnr),
    "copy 50MB (roofline ref)": lambda: xff.clone(),
}

results = {name: [] for name in variants}
for _ in range(3):  # warmup all
    for fn in variants.values():
        fn()
torch.cuda.synchronize()
for rnd in range(5):
    for name, fn in variants.items():
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            fn()
        torch.cuda.synchronize()
        results[name].append((time.perf_counter() - t0) / 10 * 1e6)

print(f"{'kernel':<34}{'median_us':>10}{'min_us':>9}")
for name, ts in results.items():
    ts.sort()
    print(f"{name:<34}{ts[len(ts)//2]:>10.1f}{ts[0]:>9.1f}")
