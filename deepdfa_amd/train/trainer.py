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

MI355X-native execution: one process per GPU (torchrun) with the RCCL DDP
engine — rank-consistent per-epoch undersampling via a shared-seed shuffle
+ strided shard (datamodule), per-step bucketed gradient all-reduce
(parallel/ddp.py), metric-counter + loss all-reduce at epoch boundaries,
rank-0 checkpoint writes; bf16 autocast; hipGraph step capture for the
launch-bound flow-GNN (capture.py; torch.cuda.CUDAGraph == hipGraph on
ROCm) when graph_capture is enabled on a single-GPU run.
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

from ..parallel.ddp import DDPEngine, init_distributed, world_size

logger = logging.getLogger(__name__)


def _dist_rank_ws():
    import torch.distributed as dist

    if dist.is_available() and dist.is_initialized():
        return dist.get_rank(), dist.get_world_size()
    return 0, 1


def _allreduce_sums(vals, device):
    """Sum a list of python floats across ranks (no-op if not distributed)."""
    import torch.distributed as dist

    if not (dist.is_available() and dist.is_initialized()):
        return vals
    t = torch.tensor(vals, dtype=torch.float64,
                     device=device if device.type == "cuda" else "cpu")
    dist.all_reduce(t)
    return t.tolist()


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
        graph_capture: bool = True,  # hipGraph step capture (GPU, ws=1, FlatAdamW)
        accumulate_grad_batches: int = 1,  # Lightning-parity micro-batching
    ):
        self.graph_capture = graph_capture
        # reference config_default.yaml:37 trainer.detect_anomaly — autograd
        # NaN/inf detection; the kernel-level analog is scripts/run_sanitize.sh
        self.detect_anomaly = detect_anomaly
        self.precision = precision
        self.max_epochs = max_epochs
        self.root = default_root_dir
        self.check_val_every = check_val_every_n_epoch
        self.periodic_every = periodic_every
        self.grad_clip = grad_clip
        self.accumulate_grad_batches = max(1, int(accumulate_grad_batches))
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

    def fit(self, model, datamodule, optimizer=None, hpo_reporter=None,
            ckpt_path: Optional[str] = None) -> Dict:
        from ..utils.logging import ScalarLogger

        rank = init_distributed()
        ws = world_size()
        if ws > 1 and torch.cuda.is_available():
            self.device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", 0)))
            torch.cuda.set_device(self.device)
        torch.manual_seed(self.seed)
        if self.detect_anomaly:
            torch.autograd.set_detect_anomaly(True)
        model = model.to(self.device)
        if optimizer is None:
            optimizer = torch.optim.Adam(model.parameters(), lr=1e-3, weight_decay=1e-2)
        if hasattr(optimizer, "flat_p") and (
            optimizer.flat_p.device.type != self.device.type
            or (self.device.index is not None
                and optimizer.flat_p.device.index != self.device.index)
        ):
            raise RuntimeError(
                "FlatAdamW was built before the model moved to "
                f"{self.device}: its flat master buffer lives on "
                f"{optimizer.flat_p.device} and the parameter views are "
                "broken. Construct it AFTER model.to(device)."
            )
        # gradient averaging: flat optimizers (FlatAdamW) do ONE flat
        # all-reduce after backward; otherwise the bucketed-overlap engine
        flat = hasattr(optimizer, "allreduce_grads")
        ddp = None
        if flat:
            if ws > 1:
                with torch.no_grad():
                    import torch.distributed as dist

                    dist.broadcast(optimizer.flat_p, src=0)
        else:
            ddp = DDPEngine(model)
        captured = None
        if (
            self.graph_capture
            and self.accumulate_grad_batches == 1
            and ws == 1
            and self.device.type == "cuda"
            and self.precision == "bf16"
            and hasattr(optimizer, "flat_g")
            and hasattr(model, "training_step_masked")
        ):
            from .capture import CapturedTrainStep

            captured = CapturedTrainStep(
                model, optimizer, batch_size=datamodule.batch_size,
                grad_clip=self.grad_clip,
            )
            logger.info("fit: hipGraph step capture enabled")
        history = []
        scalars = ScalarLogger(self.root) if rank == 0 else None
        # mid-training resume (Lightning fit(ckpt_path=...) semantics,
        # SURVEY §5.4): restore weights + optimizer state + epoch/step
        # counters and continue from the NEXT epoch
        start_epoch = 0
        if ckpt_path:
            payload = self.load_checkpoint(model, ckpt_path, optimizer)
            start_epoch = int(payload.get("epoch", -1)) + 1
            self.global_step = int(payload.get("global_step", 0))
            # same-run-dir resume: restore the best-val watermark from the
            # existing performance-*.ckpt names so a worse post-resume epoch
            # doesn't displace the historical best (save_top_k pruning)
            bp = self.best_checkpoint()
            if bp:
                m = re.search(r"performance-\d+-\d+-([\d.]+)\.ckpt$", bp)
                if m:
                    self._best_path = bp
                    self._best_val = float(m.group(1))
            if rank == 0:
                logger.info("fit: resumed from %s at epoch %d (step %d)",
                            ckpt_path, start_epoch, self.global_step)
        # one shared generator, advanced once per epoch on EVERY rank: all
        # ranks draw the identical epoch permutation, then take their shard.
        # On resume, replay the completed epochs' loader draws (index-only,
        # no graph loading) so the undersample state and permutation stream
        # match an uninterrupted run.
        gen = torch.Generator().manual_seed(self.seed)
        for _ in range(start_epoch):
            datamodule.train_dataloader(generator=gen, rank=rank, world=ws)
        for epoch in range(start_epoch, self.max_epochs):
            model.train()
            t0 = _time.perf_counter()
            train_loader = datamodule.train_dataloader(generator=gen, rank=rank, world=ws)
            n_batches = 0
            loss_sum = 0.0
            accum = self.accumulate_grad_batches
            # trailing-group detection needs the batch count; DataLoader
            # knows it without materializing the epoch
            total_b = len(train_loader) if accum > 1 else None
            for bi, batch in enumerate(train_loader):
                if captured is not None:
                    # padding happens on the CPU batch; the captured step
                    # copies the padded tensors into its static device buffers
                    captured(batch)
                else:
                    # accumulation (Lightning accumulate_grad_batches
                    # semantics): zero at group start, loss/accum per
                    # micro-batch, one averaging/clip/step at group end
                    # (also handles a short trailing group)
                    group_start = bi % accum == 0
                    group_end = (bi + 1) % accum == 0 or (
                        total_b is not None and bi + 1 == total_b
                    )
                    batch = self._to_device(batch)
                    if group_start:
                        optimizer.zero_grad(set_to_none=False)
                    with self._autocast():
                        loss = model.training_step(batch)
                    scaled = loss / accum if accum > 1 else loss
                    if ddp is not None and not group_end:
                        # hook engine: suppress per-micro-batch bucket
                        # all-reduces; grads accumulate locally until the
                        # group-end finalize (parallel/ddp.py no_sync)
                        with ddp.no_sync():
                            scaled.backward()
                    else:
                        scaled.backward()
                    if group_end:
                        if flat and ws > 1:
                            optimizer.allreduce_grads()
                        elif ddp is not None:
                            ddp.finalize()
                        if self.grad_clip:
                            torch.nn.utils.clip_grad_norm_(model.parameters(), self.grad_clip)
                        optimizer.step()
                    loss_sum += float(loss.detach())
                self.global_step += 1
                n_batches += 1
            if captured is not None:
                loss_sum = captured.pop_loss_sum()
            train_metrics = model.epoch_metrics("train")  # counter all-reduce
            loss_sum, nb = _allreduce_sums([loss_sum, float(n_batches)], self.device)
            row = {
                "epoch": epoch,
                "train_loss": loss_sum / max(1.0, nb),
                "epoch_time_s": _time.perf_counter() - t0,
                **train_metrics,
            }
            if (epoch + 1) % self.check_val_every == 0:
                val = self.validate(model, datamodule, _during_fit=True)
                row.update(val)
                val_loss = val["val_loss"]  # identical on every rank (reduced)
                if val_loss < self._best_val:
                    self._best_val = val_loss
                    if rank == 0:
                        # remove previous best (save_top_k: 1)
                        if self._best_path and os.path.exists(self._best_path):
                            os.unlink(self._best_path)
                    name = f"performance-{epoch:02d}-{self.global_step:02d}-{val_loss:02f}.ckpt"
                    self._best_path = os.path.join(self.ckpt_dir, name)
                    if rank == 0:
                        self.save_checkpoint(model, optimizer, epoch, val_loss, name)
                if rank == 0:
                    self.save_checkpoint(model, optimizer, epoch, val_loss, "last.ckpt")
            if (epoch + 1) % self.periodic_every == 0 and rank == 0:
                self.save_checkpoint(
                    model,
                    optimizer,
                    epoch,
                    row.get("val_loss", float("nan")),
                    f"periodical-{epoch}-{self.global_step}.ckpt",
                )
            history.append(row)
            if scalars is not None:
                scalars.log({**row, "lr": optimizer.param_groups[0].get("lr", 0.0)}, self.global_step)
            if hpo_reporter is not None and "val_f1" in row:
                hpo_reporter.report_intermediate(row["val_f1"])
            logger.info("epoch %d: %s", epoch, json.dumps({k: round(v, 5) if isinstance(v, float) else v for k, v in row.items()}))
        if scalars is not None:
            scalars.close()
        if captured is not None:
            logger.info(
                "fit: %d steps captured-replayed, %d eager fallbacks, %d shape buckets",
                captured.steps - captured.eager_steps, captured.eager_steps,
                len(captured.buckets),
            )
        if ws > 1:
            import torch.distributed as dist

            dist.barrier()  # rank0 finished writing checkpoints
        return {"history": history, "best_checkpoint": self.best_checkpoint()}

    @torch.no_grad()
    def validate(self, model, datamodule, _during_fit: bool = False) -> Dict:
        rank, ws = _dist_rank_ws()
        model = model.to(self.device)
        model.eval()
        loss_sum, n = 0.0, 0
        for batch in datamodule.val_dataloader(rank=rank, world=ws):
            batch = self._to_device(batch)
            with self._autocast():
                loss = model.validation_step(batch)
            loss_sum += float(loss)
            n += 1
        loss_sum, nf = _allreduce_sums([loss_sum, float(n)], self.device)
        out = {"val_loss": loss_sum / max(1.0, nf)}
        out.update(model.epoch_metrics("val"))  # counter all-reduce inside
        if not _during_fit:
            model.train()
        return out

    @torch.no_grad()
    def test(self, model, datamodule, out_dir: Optional[str] = None) -> Dict:
        rank, ws = _dist_rank_ws()
        model = model.to(self.device)
        model.eval()
        loss_sum, n = 0.0, 0
        for batch in datamodule.test_dataloader(rank=rank, world=ws):
            batch = self._to_device(batch)
            with self._autocast():
                loss = model.test_step(batch)
            loss_sum += float(loss)
            n += 1
        loss_sum, nf = _allreduce_sums([loss_sum, float(n)], self.device)
        if ws > 1:
            # gather per-example preds/labels (already on CPU) so the PR
            # curve / classification report cover the whole test split
            import torch.distributed as dist

            gp: list = [None] * ws
            gl: list = [None] * ws
            dist.all_gather_object(gp, model._test_preds)
            dist.all_gather_object(gl, model._test_labels)
            model._test_preds = [t for lst in gp for t in lst]
            model._test_labels = [t for lst in gl for t in lst]
        results = {"test_loss": loss_sum / max(1.0, nf)}
        results.update(
            model.test_epoch_end(out_dir=(out_dir or self.root) if rank == 0 else None)
        )
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
