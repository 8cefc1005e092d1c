"""Time flash fwd/bwd kernels with vs without the (H,L,L) fp32 bias."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepdfa_amd.ops import load_ext
ext = load_ext(required=True)
dev = "cuda"
torch.manual_seed(0)

def t(fn, iters=100):
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    for _ in range(10): fn()
    torch.cuda.synchronize(); s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000

for B in (8, 16):
    L, H = 512, 12
    D = H * 64
    q = torch.randn(B, L, D, device=dev, dtype=torch.bfloat16)
    k = torch.randn(B, L, D, device=dev, dtype=torch.bfloat16)
    v = torch.randn(B, L, D, device=dev, dtype=torch.bfloat16)
    bias = torch.randn(H, L, L, device=dev, dtype=torch.float32)
    O, lse = ext.flash_attn_fwd(q, k, v, H, None, bias, 1.0, False, 0.0, 0)
    dO = torch.randn_like(O)
    for name, b in (("bias", bias), ("none", None)):
        fwd = t(lambda: ext.flash_attn_fwd(q, k, v, H, None, b, 1.0, False, 0.0, 0))
        bwd = t(lambda: ext.flash_attn_bwd(dO, q, k, v, O, lse, H, None, b, 1.0,
                                           False, 0.0, 0, False, False, None))
        print(f"B={B} {name}: fwd {fwd:7.2f} us   bwd(dterm+dq+dkv) {bwd:7.2f} us")
