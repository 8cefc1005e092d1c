"""Diagnose hipBLASLt GEMM times for the flow-GNN's shapes on MI355X.

rocprof showed every torch.matmul/addmm bf16 call in the training step
taking 84-91 us regardless of shape (even 256x256x256). This probe times
each shape in isolation, with and without bias, under the default backend
and under ROCBLAS (TORCH_BLAS_PREFER_HIPBLASLT=0 must be set before launch)
to locate the pathology. Run on a GPU box:
  python tools/gemm_probe.py
  TORCH_BLAS_PREFER_HIPBLASLT=0 python tools/gemm_probe.py
  PYTORCH_TUNABLEOP_ENABLED=1 python tools/gemm_probe.py
"""

import os
import time

import torch

assert torch.cuda.is_available()
dev = "cuda"
bf = torch.bfloat16


def t(fn, n=100):
    for _ in range(10):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e6


N = 11600
cases = {}
a = torch.randn(N, 128, device=dev, dtype=bf)
w = torch.randn(384, 128, device=dev, dtype=bf)
cases["fwd gi (N,128)@(128,384)t"] = lambda: a @ w.t()
b_ = torch.randn(384, device=dev, dtype=bf)
cases["fwd gi addmm bias"] = lambda: torch.addmm(b_, a, w.t())
g = torch.randn(N, 512, device=dev, dtype=bf)
wc = torch.randn(512, 256, device=dev, dtype=bf)
cases["bwd gradA (N,512)@(512,256)"] = lambda: g @ wc
m = torch.randn(N, 128, device=dev, dtype=bf)
gt = torch.randn(N, 384, device=dev, dtype=bf)
cases["wgrad (384,N)@(N,128) [t()]"] = lambda: gt.t() @ m
x256 = torch.randn(256, 256, device=dev, dtype=bf)
w256 = torch.randn(256, 256, device=dev, dtype=bf)
cases["mlp (256,256)@(256,256)t"] = lambda: x256 @ w256.t()
gate_w = torch.randn(1, 256, device=dev, dtype=bf)
xp = torch.randn(N, 256, device=dev, dtype=bf)
cases["gate (N,256)@(256,1)t"] = lambda: xp @ gate_w.t()
cases["gate F.linear"] = lambda: torch.nn.functional.linear(xp, gate_w)
af = torch.randn(N, 128, device=dev)
wf = torch.randn(384, 128, device=dev)
cases["fwd gi fp32"] = lambda: af @ wf.t()
big = torch.randn(4096, 4096, device=dev, dtype=bf)
cases["4096^3 bf16"] = lambda: big @ big

print(f"backend: PREFER_HIPBLASLT={os.environ.get('TORCH_BLAS_PREFER_HIPBLASLT','1')} "
      f"TUNABLEOP={os.environ.get('PYTORCH_TUNABLEOP_ENABLED','0')}")
for name, fn in cases.items():
    print(f"{name:<36}{t(fn):>10.1f} us")
