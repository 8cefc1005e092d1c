"""A/B the wgrad2 K-chunk floor (DFA_WG2_KFLOOR) on the transformer wgrad
shapes, plus an embed_scatter baseline timing."""
import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepdfa_amd.ops import load_ext
ext = load_ext(required=True)
dev = "cuda"
torch.manual_seed(0)

def t(fn, iters=200):
    s = torch.cuda.Event(True); e = torch.cuda.Event(True)
    for _ in range(20): fn()
    torch.cuda.synchronize(); s.record()
    for _ in range(iters): fn()
    e.record(); torch.cuda.synchronize()
    return s.elapsed_time(e) / iters * 1000  # us

shapes = [(4096, 768, 768), (4096, 2304, 768), (4096, 3072, 768), (4096, 768, 3072),
          (8192, 768, 768), (8192, 3072, 768)]
tens = {}
for K, M, C in shapes:
    A = torch.randn(K, M, device=dev).to(torch.bfloat16)
    B = torch.randn(K, C, device=dev).to(torch.bfloat16)
    O = torch.zeros(M, C, device=dev, dtype=torch.float32)
    tens[(K, M, C)] = (A, B, O)
for floor in (256, 384, 512, 768, 1024):
    os.environ["DFA_WG2_KFLOOR"] = str(floor)
    row = []
    for K, M, C in shapes:
        A, B, O = tens[(K, M, C)]
        us = t(lambda: ext.wgrad(A, B, out=O))
        row.append(f"{K}x{M}x{C}:{us:7.2f}")
    print(f"floor={floor:4d}  " + "  ".join(row))

dY = torch.randn(4096, 768, device=dev)
idx = torch.randint(0, 32100, (4096,), device=dev)
G = torch.zeros(32100, 768, device=dev, dtype=torch.float32)
print(f"embed_scatter f32 4096x768->32100: {t(lambda: ext.embed_scatter(dY, idx, 32100, -1, out=G)):.2f} us")
dY16 = dY.to(torch.bfloat16)
print(f"embed_scatter bf16: {t(lambda: ext.embed_scatter(dY16, idx, 32100, -1, out=G)):.2f} us")
idx2 = torch.randint(0, 50265, (8192,), device=dev)
dY2 = torch.randn(8192, 768, device=dev)
G2 = torch.zeros(50265, 768, device=dev, dtype=torch.float32)
print(f"embed_scatter f32 8192x768->50265: {t(lambda: ext.embed_scatter(dY2, idx2, 50265, -1, out=G2)):.2f} us")
