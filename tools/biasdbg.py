import sys; sys.path.insert(0, '/root/repo')
import torch
from deepdfa_amd.ops.transformer import flash_attention_qkv, fused_qkv
from deepdfa_amd.ops import load_ext
ext = load_ext(required=True)
dev = "cuda:0"
torch.manual_seed(0)
B, L, H, d = 2, 64, 2, 64
x = torch.randn(B, L, H*d, device=dev, dtype=torch.bfloat16, requires_grad=True)
wq = torch.randn(H*d, H*d, device=dev) * 0.05
qkv = fused_qkv(x, wq, wq, wq)
bias = torch.randn(H, L, L, device=dev).float()
accum = torch.zeros(H, L, L, device=dev)
out = flash_attention_qkv(qkv, H, valid=None, bias=bias, scale=1.0, causal=False,
                          dropout_p=0.0, bias_accum=accum)
out.float().pow(2).mean().backward()
print("accum sum:", float(accum.abs().sum()))
# reference: differentiable bias without accum
x2 = x.detach().requires_grad_(True)
qkv2 = fused_qkv(x2, wq, wq, wq)
bias2 = bias.detach().requires_grad_(True)
out2 = flash_attention_qkv(qkv2, H, valid=None, bias=bias2, scale=1.0, causal=False, dropout_p=0.0)
out2.float().pow(2).mean().backward()
print("ref dbias sum:", float(bias2.grad.abs().sum()))
print("match:", torch.allclose(accum, bias2.grad, rtol=0.05, atol=1e-4))
