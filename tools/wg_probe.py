import sys, os, time
sys.path.insert(0, '/root/repo')
import torch
from deepdfa_amd.ops import load_ext
ext = load_ext(required=True)
dev = "cuda"
torch.manual_seed(0)
B, D = 257, 256
x = (torch.randn(B, D, device=dev)*0.3).to(torch.bfloat16)
h1 = torch.relu(torch.randn(B, D, device=dev))
h2 = torch.relu(torch.randn(B, D, device=dev))
W1 = torch.randn(D, D, device=dev)*0.05
W2 = torch.randn(D, D, device=dev)*0.05
W3 = torch.randn(D, device=dev)*0.05
dl = torch.randn(B, device=dev)

def bench(fn, n=200):
    for _ in range(20): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/n*1e6

print("mlp3_bwd+wgrad:", bench(lambda: ext.mlp3_bwd(dl, x, h1, h2, W1, W2, W3)), "us")
