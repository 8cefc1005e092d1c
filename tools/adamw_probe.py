import sys, os, time
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepdfa_amd.parallel.optim import FlatAdamW
n = 125_000_000
p = torch.nn.Parameter(torch.randn(n, device="cuda"))
opt = FlatAdamW([p], lr=1e-4)
opt.flat_g.normal_()
def bench(k=50):
    for _ in range(5): opt.step()
    torch.cuda.synchronize(); t0 = time.perf_counter()
    for _ in range(k): opt.step()
    torch.cuda.synchronize(); return (time.perf_counter()-t0)/k*1e6
print("default:", round(bench(), 1), "us")
