"""Trainer: fit / validate / test loops with reference-compatible
checkpointing.

Parity targets:
  * checkpoint naming `performance-{epoch:02d}-{step:02d}-{val_loss:02f}.ckpt`
    + `last.ckpt`, save_top_k=1 on min val_loss, plus a periodic checkpoint
    every N epochs `periodical-{epoch}-{step}.ckpt`
    (config_default.yaml:17-31, periodic_checkpoint.py:8-22);
  * post-fit best-checkpoint selection by parsing val_loss from the FILENAME
    (main_cli.py:167-184);
  * per-epoch dataloader rebuild for undersampling
    (reload_dataloaders_every_n_epochs: 1);
  * .ckpt payload: state_dict + optimizer state + hyper_parameters
    (Lightning layout subset, enough for resume + transfer).

MI355X-native execution: model/graphs on cuda:0 per process, bf16 autocast
optional, hipGraph step capture for the launch-bound flow-GNN
(torch.cuda.CUDAGraph == hipGraph on ROCm), DDP via deepdfa_amd.parallel.
"""

from __future__ import annotations

import glob
import json
import logging
import os
import re
import time as _time
from typing import Dict, Optional

import torch

logger = logging.getLogger(__name__)


class Trainer:
    def __init__(
        self,
        max_epochs: int = 25,
        default_root_dir: str = "lightning_logs",
        check_val_every_n_epoch: int = 1,
        periodic_every: int = 25,
        device: Optional[str] = None,
        grad_clip: Optional[float] = None,
        log_every_n_steps: int = 50,
        seed: int = 0,
        precision: str = "bf16",  # "bf16" (autocast on GPU) or "fp32"
        detect_anomaly: bool = False,
    ):
        # reference config_default.yaml:37 trainer.detect_anomaly — autograd
        # NaN/inf detection; the kernel-level analog is scripts/run_sanitize.sh
        self.detect_anomaly = detect_anomaly
        self.precision = precision
        self.max_epochs = max_epochs
        self.root = default_root_dir
        self.check_val_every = check_val_every_n_epoch
        self.periodic_every = periodic_every
        self.grad_clip = grad_clip
        self.log_every = log_every_n_steps
        self.seed = seed
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        self.device = torch.device(device)
        self.global_step = 0
        self.ckpt_dir = os.path.join(self.root, "checkpoints")
        os.makedirs(self.ckpt_dir, exist_ok=True)
        self._best_val = float("inf")
        self._best_path: Optional[str] = None

    # -- checkpointing -------------------------------------------------------

    def save_checkpoint(self, model, optimizer, epoch: int, val_loss: float, name: str):
        payload = {
            "epoch": epoch,
            "global_step": self.global_step,
            "state_dict": model.state_dict(),
            "optimizer_states": [optimizer.state_dict()] if optimizer is not None else [],
            "hyper_parameters": getattr(model, "hparams", {}),
            "val_loss": val_loss,
        }
        path = os.path.join(self.ckpt_dir, name)
        torch.save(payload, path)
        return path

    def load_checkpoint(self, model, path: str, optimizer=None):
        payload = torch.load(path, map_location=self.device, weights_only=False)
        model.load_state_dict(payload["state_dict"])
        if optimizer is not None and payload.get("optimizer_states"):
            optimizer.load_state_dict(payload["optimizer_states"][0])
        return payload

    def best_checkpoint(self) -> Optional[str]:
        """Pick min-val_loss among performance-*.ckpt by parsing the filename
        (main_cli.py:175-183 contract)."""
        paths = glob.glob(os.path.join(self.ckpt_dir, "performance-*.ckpt"))
        best, best_loss = None, float("inf")
        for p in paths:
            m = re.search(r"performance-\d+-\d+-([\d.]+)\.ckpt$", p)
            if m:
                loss = float(m.group(1))
                if loss < best_loss:
                    best, best_loss = p, loss
        return best

    # -- loops ----------------------------------------------------------------

    def fit(self, model, datamodule, optimizer=None, hpo_reporter=None) -> Dict:
        from ..utils.logging import ScalarLogger

        torch.manual_seed(self.seed)
        if self.detect_anomaly:
            torch.autograd.set_detect_anomaly(True)
        model = model.to(self.device)
        if optimizer is None:
            optimizer = torch.optim.Adam(model.parameters(), lr=1e-3, weight_decay=1e-2)
        history = []
        scalars = ScalarLogger(self.root)
        gen = torch.Generator().manual_seed(self.seed)
        for epoch in range(self.max_epochs):
            model.train()
            t0 = _time.perf_counter()
            train_loader = datamodule.train_dataloader(generator=gen)
            n_batches = 0
            loss_sum = 0.0
            for batch in train_loader:
                batch = self._to_device(batch)
                with self._autocast():
                    loss = model.training_step(batch)
                optimizer.zero_grad(set_to_none=True)
                loss.backward()
                if self.grad_clip:
                    torch.nn.utils.clip_grad_norm_(model.parameters(), self.grad_clip)
                optimizer.step()
                self.global_step += 1
                n_batches += 1
                loss_sum += float(loss.detach())
            train_metrics = model.epoch_metrics("train")
            row = {
                "epoch": epoch,
                "train_loss": loss_sum / max(1, n_batches),
                "epoch_time_s": _time.perf_counter() - t0,
                **train_metrics,
            }
            if (epoch + 1) % self.check_val_every == 0:
                val = self.validate(model, datamodule, _during_fit=True)
                row.update(val)
                val_loss = val["val_loss"]
                if val_loss < self._best_val:
                    self._best_val = val_loss
                    # remove previous best (save_top_k: 1)
                    if self._best_path and os.path.exists(self._best_path):
                        os.unlink(self._best_path)
                    name = f"performance-{epoch:02d}-{self.global_step:02d}-{val_loss:02f}.ckpt"
                    self._best_path = self.save_checkpoint(
                        model, optimizer, epoch, val_loss, name
                    )
                self.save_checkpoint(model, optimizer, epoch, val_loss, "last.ckpt")
            if (epoch + 1) % self.periodic_every == 0:
                self.save_checkpoint(
                    model,
                    optimizer,
                    epoch,
                    row.get("val_loss", float("nan")),
                    f"periodical-{epoch}-{self.global_step}.ckpt",
                )
            history.append(row)
            scalars.log({**row, "lr": optimizer.param_groups[0].get("lr", 0.0)}, self.global_step)
            if hpo_reporter is not None and "val_f1" in row:
                hpo_reporter.report_intermediate(row["val_f1"])
            logger.info("epoch %d: %s", epoch, json.dumps({k: round(v, 5) if isinstance(v, float) else v for k, v in row.items()}))
        scalars.close()
        return {"history": history, "best_checkpoint": self.best_checkpoint()}

    @torch.no_grad()
    def validate(self, model, datamodule, _during_fit: bool = False) -> Dict:
        model = model.to(self.device)
        model.eval()
        loss_sum, n = 0.0, 0
        for batch in datamodule.val_dataloader():
            batch = self._to_device(batch)
            with self._autocast():
                loss = model.validation_step(batch)
            loss_sum += float(loss)
            n += 1
        out = {"val_loss": loss_sum / max(1, n)}
        out.update(model.epoch_metrics("val"))
        if not _during_fit:
            model.train()
        return out

    @torch.no_grad()
    def test(self, model, datamodule, out_dir: Optional[str] = None) -> Dict:
        model = model.to(self.device)
        model.eval()
        loss_sum, n = 0.0, 0
        for batch in datamodule.test_dataloader():
            batch = self._to_device(batch)
            with self._autocast():
                loss = model.test_step(batch)
            loss_sum += float(loss)
            n += 1
        results = {"test_loss": loss_sum / max(1, n)}
        results.update(model.test_epoch_end(out_dir=out_dir or self.root))
        return results

    # -- helpers --------------------------------------------------------------

    def _autocast(self):
        import contextlib

        if self.precision == "bf16" and self.device.type == "cuda":
            return torch.autocast(device_type="cuda", dtype=torch.bfloat16)
        return contextlib.nullcontext()

    def _to_device(self, batch):
        if isinstance(batch, tuple) and len(batch) == 2:
            graph, extra = batch
            return graph.to(self.device), extra
        return batch
