import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, yaml
from deepdfa_amd.train.main_cli import parse_cli, build
args, cfg = parse_cli(["fit", "--config", "configs/config_bigvul.yaml",
                       "--config", "configs/config_ggnn.yaml",
                       "--trainer.max_epochs", "1", "--data.n_synthetic", "1000"])
dm, model, trainer = build(cfg)
from deepdfa_amd.parallel.optim import FlatAdamW
model = model.to("cuda")
opt = FlatAdamW([p for p in model.parameters() if p.requires_grad], l2_mode=True, **cfg["optimizer"])
from deepdfa_amd.train.capture import CapturedTrainStep
cap = CapturedTrainStep(model, opt, batch_size=dm.batch_size, grad_clip=trainer.grad_clip)
gen = torch.Generator().manual_seed(0)
loader = dm.train_dataloader(generator=gen)
batch = next(iter(loader))
g, extra = batch
print("batch:", g.num_graphs, g.num_nodes, g.num_edges)
for k, v in g.ndata.items():
    print(" ndata", k, v.dtype, v.shape, v.is_contiguous())
from deepdfa_amd.graph.pad import bucket_shape, pad_batch
shape = bucket_shape(g, dm.batch_size + 1)
padded, w = pad_batch(g, *shape)
sg = padded.to("cuda")
for k, v in sg.ndata.items():
    print(" static ndata", k, v.device, v.is_contiguous())
idx = torch.stack([sg.ndata[f"_ABS_DATAFLOW_{of}"] for of in ("api","datatype","literal","operator")], dim=1)
print("idx:", idx.device, idx.dtype, idx.is_contiguous())
try:
    cap(batch)
    print("capture OK")
except Exception as e:
    import traceback; traceback.print_exc()
