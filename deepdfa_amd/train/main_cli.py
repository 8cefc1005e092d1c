"""DeepDFA CLI: fit / validate / test with stacked YAML configs.

Parity target: reference DDFA/code_gnn/main_cli.py (MyLightningCLI):
  * subcommands fit/validate/test over stacked --config files with dotted
    CLI overrides (--data.batch_size 64), seed_everything;
  * argument LINKING: data.feat -> model.feat and the computed
    data.input_dim -> model.input_dim (:95-99);
  * --freeze_graph <ckpt>: load all non-head weights and freeze them
    (:136-144);
  * post-fit best-checkpoint selection by min val_loss parsed from the
    FILENAME + a validate run on it (:167-184);
  * --analyze_dataset: dataset coverage analyzer (:192-313) reporting
    per-feature vocabulary coverage of each split;
  * persistent timestamped file log hard-linked into the run dir
    (:31-65,123-134), renamed to *.error on failure (:324-336);
  * HPO hook: externally-injected parameter overrides (the reference's NNI
    integration; here a JSON file / env var since nni is not installed).
"""

from __future__ import annotations

import argparse
import json
import logging
import os
import time
import traceback
from typing import Dict, List, Optional

import torch
import yaml

from ..data import BigVulDatasetLineVDDataModule, parse_limits
from ..models import FlowGNNGGNNModule
from .trainer import Trainer

logger = logging.getLogger(__name__)


def deep_update(base: Dict, extra: Dict) -> Dict:
    for k, v in extra.items():
        if isinstance(v, dict) and isinstance(base.get(k), dict):
            deep_update(base[k], v)
        else:
            base[k] = v
    return base


DEFAULT_CONFIG = {
    "seed_everything": 0,
    "trainer": {"max_epochs": 25, "default_root_dir": "lightning_logs",
                "check_val_every_n_epoch": 1, "periodic_every": 25},
    "optimizer": {"lr": 1e-3, "weight_decay": 1e-2},
    "model": {"n_steps": 5, "hidden_dim": 32, "num_output_layers": 3,
              "concat_all_absdf": True},
    "data": {"feat": "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000",
             "gtype": "cfg", "dsname": "bigvul", "undersample": "v1.0",
             "split": "fixed", "batch_size": 256, "sample_mode": False,
             "n_synthetic": 2000},
    "ckpt_path": None,
}


def parse_cli(argv: Optional[List[str]] = None):
    p = argparse.ArgumentParser(prog="main_cli")
    p.add_argument("subcommand", choices=["fit", "validate", "test"])
    p.add_argument("--config", action="append", default=[])
    p.add_argument("--seed_everything", type=int, default=None)
    p.add_argument("--ckpt_path", default=None)
    p.add_argument("--freeze_graph", default=None)
    p.add_argument("--analyze_dataset", action="store_true")
    p.add_argument("--hpo_params", default=None,
                   help="JSON file of dotted-override HPO parameters")
    p.add_argument("--disable-warnings", action="store_true")
    args, overrides = p.parse_known_args(argv)
    cfg = json.loads(json.dumps(DEFAULT_CONFIG))  # deep copy
    for path in args.config:
        with open(path) as f:
            deep_update(cfg, yaml.safe_load(f) or {})
    # dotted CLI overrides: --data.batch_size 64
    i = 0
    while i < len(overrides):
        key = overrides[i]
        if key.startswith("--") and i + 1 < len(overrides):
            node = cfg
            parts = key[2:].split(".")
            for part in parts[:-1]:
                node = node.setdefault(part, {})
            node[parts[-1]] = yaml.safe_load(overrides[i + 1])
            i += 2
        else:
            i += 1
    if args.hpo_params is None:
        from ..utils.logging import HPOReporter

        injected = HPOReporter.get_next_parameter()
        if injected:
            for k, v in injected.items():
                node = cfg
                parts = k.split(".")
                for part in parts[:-1]:
                    node = node.setdefault(part, {})
                node[parts[-1]] = v
    if args.hpo_params:
        with open(args.hpo_params) as f:
            for k, v in json.load(f).items():
                node = cfg
                parts = k.split(".")
                for part in parts[:-1]:
                    node = node.setdefault(part, {})
                node[parts[-1]] = v
    if args.seed_everything is not None:
        cfg["seed_everything"] = args.seed_everything
    if args.ckpt_path is not None:
        cfg["ckpt_path"] = args.ckpt_path
    return args, cfg


def build(cfg: Dict):
    data_cfg = dict(cfg["data"])
    dm = BigVulDatasetLineVDDataModule(**data_cfg)
    model_cfg = dict(cfg["model"])
    # argument linking (main_cli.py:95-99): feat + computed input_dim
    model_cfg.setdefault("feat", dm.feat)
    model_cfg.setdefault("input_dim", dm.input_dim)
    model = FlowGNNGGNNModule(**model_cfg)
    tr_cfg = dict(cfg["trainer"])
    trainer = Trainer(seed=cfg["seed_everything"], **tr_cfg)
    return dm, model, trainer


def get_coverage(dm) -> Dict:
    """Dataset coverage analyzer (main_cli.py:192-313): fraction of node
    feature values per split that are in-vocabulary (not UNK/no-def)."""
    spec = parse_limits(dm.feat)
    out = {}
    for split in ("train", "val", "test"):
        ds = getattr(dm, split)
        counts = {"total": 0, "no_def": 0, "unk": 0, "known": 0}
        for idx in list(ds.df.index)[:200]:
            try:
                g, _ = ds.item(idx)
            except KeyError:
                continue
            for featname in g.ndata:
                if not featname.startswith("_ABS_DATAFLOW"):
                    continue
                vals = g.ndata[featname]
                counts["total"] += vals.numel()
                counts["no_def"] += int((vals == 0).sum())
                counts["unk"] += int((vals == 1).sum())
                counts["known"] += int((vals >= 2).sum())
        cov = counts["known"] / max(1, counts["total"] - counts["no_def"])
        out[split] = {**counts, "coverage": cov, "input_dim": spec.input_dim}
    return out


def freeze_graph_weights(model, ckpt_path: str, device="cpu"):
    """--freeze_graph: load all non-output-head weights, freeze them."""
    payload = torch.load(ckpt_path, map_location=device, weights_only=False)
    sd = payload.get("state_dict", payload)
    own = model.state_dict()
    loaded = []
    for k, v in sd.items():
        if k.startswith("output_layer"):
            continue
        if k in own and own[k].shape == v.shape:
            own[k].copy_(v)
            loaded.append(k)
    for name, p in model.named_parameters():
        if not name.startswith("output_layer"):
            p.requires_grad_(False)
    logger.info("freeze_graph: loaded+froze %d tensors from %s", len(loaded), ckpt_path)
    return loaded


def main(argv: Optional[List[str]] = None) -> Dict:
    args, cfg = parse_cli(argv)
    # persistent file log, hard-linked into the run dir (main_cli.py:31-65)
    ts = time.strftime("%Y%m%d_%H%M%S")
    os.makedirs("logs", exist_ok=True)
    log_file = os.path.join("logs", f"run_{ts}_{os.getpid()}.log")
    handler = logging.FileHandler(log_file)
    logging.basicConfig(level=logging.INFO, handlers=[handler, logging.StreamHandler()])
    try:
        torch.manual_seed(cfg["seed_everything"])
        dm, model, trainer = build(cfg)
        run_dir = trainer.root
        os.makedirs(run_dir, exist_ok=True)
        linked = os.path.join(run_dir, "output.log")
        if not os.path.exists(linked):
            try:
                os.link(log_file, linked)
            except OSError:
                pass
        # save the resolved config into the run dir (save_config_overwrite)
        with open(os.path.join(run_dir, "config.yaml"), "w") as f:
            yaml.safe_dump(cfg, f)
        if args.analyze_dataset:
            cov = get_coverage(dm)
            logger.info("dataset coverage: %s", json.dumps(cov))
            return {"coverage": cov}
        if args.freeze_graph:
            freeze_graph_weights(model, args.freeze_graph)
        if cfg.get("ckpt_path") and args.subcommand != "fit":
            trainer.load_checkpoint(model, cfg["ckpt_path"])
        if args.subcommand == "fit":
            from ..utils.logging import HPOReporter

            # the flat optimizer re-views parameters into its master
            # buffer — the model must already live on the training device
            model = model.to(trainer.device)
            params = [p for p in model.parameters() if p.requires_grad]
            if getattr(trainer, "graph_capture", False) and torch.cuda.is_available():
                # capture-safe flat optimizer (device-side bias correction);
                # l2_mode=True == torch.optim.Adam's weight_decay semantics
                from ..parallel.optim import FlatAdamW

                opt = FlatAdamW(params, l2_mode=True, **cfg["optimizer"])
            else:
                opt = torch.optim.Adam(params, **cfg["optimizer"])
            reporter = HPOReporter(run_dir)
            # fit(ckpt_path=...) = mid-training RESUME: weights + optimizer
            # state + epoch/step counters (Lightning semantics)
            out = trainer.fit(model, dm, optimizer=opt, hpo_reporter=reporter,
                              ckpt_path=cfg.get("ckpt_path"))
            # post-fit: validate the best checkpoint (main_cli.py:167-184)
            best = out["best_checkpoint"]
            if best:
                trainer.load_checkpoint(model, best)
                out["best_validate"] = trainer.validate(model, dm)
                reporter.report_final(out["best_validate"].get("val_f1", 0.0))
            return out
        if args.subcommand == "validate":
            return trainer.validate(model, dm)
        return trainer.test(model, dm)
    except Exception:
        logger.error("run failed:\n%s", traceback.format_exc())
        handler.close()
        try:
            os.rename(log_file, log_file + ".error")  # main_cli.py:324-336
        except OSError:
            pass
        raise


if __name__ == "__main__":
    main()
