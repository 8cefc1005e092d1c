import sys; sys.path.insert(0, "/root/repo")
import torch
from deepdfa_amd.graph.synthetic import synthetic_cfg_batch
from deepdfa_amd.models import FlowGNNGGNNModule
from deepdfa_amd.parallel.optim import FlatAdamW
torch.manual_seed(0)
model = FlowGNNGGNNModule(input_dim=1002).to("cuda")
g = synthetic_cfg_batch(48, seed=7).to("cuda")
opt = FlatAdamW(model.parameters(), lr=1e-3)
def loss_fn():
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        return model.training_step((g, {}))
opt.zero_grad(); loss_fn().backward()
once = {n: p.grad.detach().clone() for n, p in model.named_parameters()}
opt.zero_grad(); loss_fn().backward(); loss_fn().backward()
for n, p in model.named_parameters():
    o, t = once[n].float(), p.grad.float()
    on = float(o.abs().max())
    ratio = float((t.abs().sum() / o.abs().sum().clamp(min=1e-12)))
    err = float((t - 2 * o).abs().max())
    flag = "BAD" if err > 1e-3 * max(on, 1e-8) else "ok "
    print(f"{flag} {n:55s} ratio={ratio:6.3f} err={err:.3e} max={on:.3e}")
