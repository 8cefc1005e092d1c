"""Flash fwd segment breakdown (DFA_FWD_PROF=1 instrumented kernel)."""
import os, sys, time
os.environ["DFA_FWD_PROF"] = "1"
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepdfa_amd.ops import load_ext
ext = load_ext(required=True)
dev, bf = "cuda", torch.bfloat16
torch.manual_seed(0)
B, L, H = 16, 512, 12
q = torch.randn(B, L, H*64, device=dev, dtype=bf)*0.3
k = torch.randn(B, L, H*64, device=dev, dtype=bf)*0.3
v = torch.randn(B, L, H*64, device=dev, dtype=bf)*0.3
valid = torch.full((B,), L, dtype=torch.int32, device=dev)
for _ in range(20):
    O, lse = ext.flash_attn_fwd(q, k, v, H, valid, None, 0.125, False, 0.0, 0)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(100):
    O, lse = ext.flash_attn_fwd(q, k, v, H, valid, None, 0.125, False, 0.0, 0)
torch.cuda.synchronize()
print("fwd us (instrumented):", (time.perf_counter()-t0)/100*1e6)
seg = ext.fwd_prof().tolist()
names = ["loop-head","kv-load-issue","QK-mfma+kfrag","softmax valu/shfl",
         "dropout+O-rescale","P-exchange shfl","vfrag+PV-mfma","epilogue"]
tot = sum(seg)
for n, s in zip(names, seg):
    print(f"{n:22s} {s:>14d} {100*s/max(1,tot):5.1f}%")
# uninstrumented timing
os.environ.pop("DFA_FWD_PROF")
import importlib
# separate timing via sdpa comparison
qh = q.view(B,L,H,64).transpose(1,2).contiguous()
kh = k.view(B,L,H,64).transpose(1,2).contiguous()
vh = v.view(B,L,H,64).transpose(1,2).contiguous()
for _ in range(10):
    o2 = torch.nn.functional.scaled_dot_product_attention(qh, kh, vh, scale=0.125)
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(100):
    o2 = torch.nn.functional.scaled_dot_product_attention(qh, kh, vh, scale=0.125)
torch.cuda.synchronize()
print("sdpa us:", (time.perf_counter()-t0)/100*1e6)
