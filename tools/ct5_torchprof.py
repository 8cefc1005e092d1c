import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
from deepdfa_amd.models.codet5 import DefectModel
from deepdfa_amd.models.t5 import T5Config
from deepdfa_amd.parallel.optim import FlatAdamW
torch.manual_seed(0)
dev = "cuda"
cfg = T5Config()
model = DefectModel(config=cfg).to(dev)
opt = FlatAdamW(model.parameters(), lr=2e-5)
b, s = 8, 512
ids = torch.randint(3, cfg.vocab_size, (b, s), device=dev)
ids[:, -1] = cfg.eos_token_id
labels = torch.randint(0, 2, (b,), device=dev)
def step():
    with torch.autocast(device_type="cuda", dtype=torch.bfloat16):
        loss, _ = model(ids, labels=labels)
    opt.zero_grad(); loss.backward(); opt.clip_grad_norm_(1.0); opt.step()
for _ in range(5): step()
torch.cuda.synchronize()
from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA], with_stack=True) as prof:
    for _ in range(3): step()
    torch.cuda.synchronize()
rows = []
for ev in prof.key_averages(group_by_stack_n=14):
    if ev.key in ("aten::copy_", "aten::contiguous", "aten::clone", "aten::cat", "aten::_to_copy", "aten::to"):
        st = [f for f in (ev.stack or []) if "deepdfa" in f or "t5.py" in f]
        rows.append((ev.self_device_time_total, ev.count, ev.key, st[:4]))
rows.sort(reverse=True)
os.makedirs("gpurun_out", exist_ok=True)
with open("gpurun_out/ct5_copies.txt", "w") as f:
    for t, c, k, st in rows[:25]:
        f.write(f"{t/1000:.3f}ms n={c} {k}\n")
        for fr in st: f.write(f"    {fr}\n")
print(open("gpurun_out/ct5_copies.txt").read())
