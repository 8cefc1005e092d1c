import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, traceback
from deepdfa_amd.train.main_cli import parse_cli, build
args, cfg = parse_cli(["fit", "--config", "configs/config_bigvul.yaml",
                       "--config", "configs/config_ggnn.yaml",
                       "--trainer.max_epochs", "1", "--data.n_synthetic", "1000"])
torch.manual_seed(cfg["seed_everything"])
dm, model, trainer = build(cfg)
from deepdfa_amd.parallel.optim import FlatAdamW
params = [p for p in model.parameters() if p.requires_grad]
opt = FlatAdamW(params, l2_mode=True, **cfg["optimizer"])
try:
    out = trainer.fit(model, dm, optimizer=opt)
    print("fit OK", [r.get("train_loss") for r in out["history"]])
except Exception:
    traceback.print_exc()
