import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, time
from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
from deepdfa_amd.models import FlowGNNGGNNModule
from deepdfa_amd.ops.flowgnn import gate_pool, mlp3
from deepdfa_amd.ops import attn_pool

def bench(fn, n=100):
    for _ in range(10): fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n): fn()
    torch.cuda.synchronize()
    return (time.perf_counter()-t0)/n*1e6

dev = torch.device("cuda:0")
torch.manual_seed(0)
model = FlowGNNGGNNModule(input_dim=1002, hidden_dim=32, n_steps=5, num_output_layers=3).to(dev)
g = synthetic_cfg_batch(256, seed=1).to(dev)
N = g.num_nodes
x1 = torch.randn(N,128,device=dev,dtype=torch.bfloat16)
x2 = torch.randn(N,128,device=dev,dtype=torch.bfloat16)
gate_nn = model.pooling.gate_nn
seq = model.output_layer

# fwd-only probes
x1r, x2r = x1.clone().requires_grad_(True), x2.clone().requires_grad_(True)
print("gate_pool fwd:", bench(lambda: gate_pool(x1, x2, gate_nn, g)), "us")
def eager_pool():
    cat = torch.cat([x1, x2], -1)
    return attn_pool(cat, gate_nn(cat.float()).squeeze(-1).to(torch.bfloat16), g)
print("eager pool fwd:", bench(eager_pool), "us")
pooled = gate_pool(x1, x2, gate_nn, g).detach()
print("mlp3 fwd:", bench(lambda: mlp3(pooled, seq[0], seq[2], seq[4])), "us")
print("eager mlp fwd:", bench(lambda: seq(pooled.float())), "us")

# fwd+bwd probes
def fused_fb():
    a = x1.detach().requires_grad_(True); b = x2.detach().requires_grad_(True)
    out = gate_pool(a, b, gate_nn, g)
    logits = mlp3(out, seq[0], seq[2], seq[4])
    logits.sum().backward()
print("fused head f+b:", bench(fused_fb), "us")
def eager_fb():
    a = x1.detach().requires_grad_(True); b = x2.detach().requires_grad_(True)
    cat = torch.cat([a, b], -1)
    out = attn_pool(cat, gate_nn(cat.float()).squeeze(-1).to(torch.bfloat16), g)
    logits = seq(out.float()).squeeze(-1)
    logits.sum().backward()
print("eager head f+b:", bench(eager_fb), "us")

# isolate mlp3 backward (wgrad included)
pooled_g = pooled.detach().requires_grad_(True)
def mlp3_bwd_only():
    logits = mlp3(pooled_g, seq[0], seq[2], seq[4])
    logits.sum().backward()
print("mlp3 f+b:", bench(mlp3_bwd_only), "us")
