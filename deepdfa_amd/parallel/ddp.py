"""Bucketed data-parallel gradient engine over RCCL/xGMI.

MI355X-native replacement for the reference's torch.nn.DataParallel
(linevul_main.py:166) and its vestigial NCCL path (run_defect.py:143-147):
one process per GPU, RCCL ("nccl" backend on ROCm) over xGMI, with

  * an initial parameter broadcast from rank 0,
  * gradient bucketing with backward overlap: per-parameter autograd hooks
    fill flat buckets in reverse parameter order; a bucket launches its
    async all_reduce the moment its last gradient lands, so communication
    of early buckets overlaps the remaining backward compute;
  * large default bucket size (64 MiB): one MI355X node's xGMI links are
    point-to-point (7 x ~153 GB/s per GPU), so per-message latency is
    amortized with few, large collectives (SURVEY.md §2.6 xGMI note) —
    288 GB HBM makes flat fp32 buckets free;
  * gradient averaging by world size (pre-divide, reduce SUM).

For the tiny flow-GNN (~1.1M params) everything lands in one bucket and
overlap is a no-op by design (fuse-into-one-bucket is the right call per
SURVEY.md §5.8 "latency-bound" note).
"""

from __future__ import annotations

import contextlib
import os
from typing import Dict, List, Optional

import torch
import torch.distributed as dist


def init_distributed(backend: Optional[str] = None) -> int:
    """Initialize from torchrun env vars. Returns rank (0 if not distributed)."""
    if not (dist.is_available() and "WORLD_SIZE" in os.environ and int(os.environ["WORLD_SIZE"]) > 1):
        return 0
    if not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        dist.init_process_group(backend=backend)
    if torch.cuda.is_available():
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
    return dist.get_rank()


def world_size() -> int:
    return dist.get_world_size() if dist.is_available() and dist.is_initialized() else 1


class _Bucket:
    def __init__(self, params: List[torch.Tensor], dtype, device):
        self.params = params
        self.numels = [p.numel() for p in params]
        self.offsets = []
        off = 0
        for n in self.numels:
            self.offsets.append(off)
            off += n
        self.flat = torch.zeros(off, dtype=dtype, device=device)
        self.views = [
            self.flat[o : o + n].view(p.shape)
            for p, o, n in zip(params, self.offsets, self.numels)
        ]
        self.filled = [False] * len(params)
        self.work = None

    @property
    def pending(self):
        return sum(1 for f in self.filled if not f)


class DDPEngine:
    """Attach to a model whose parameters are replicated across ranks."""

    def __init__(
        self,
        model: torch.nn.Module,
        bucket_cap_mb: float = 64.0,
        grad_dtype: Optional[torch.dtype] = None,
    ):
        self.model = model
        self.enabled = world_size() > 1
        self.params = [p for p in model.parameters() if p.requires_grad]
        self._sync = True
        if not self.enabled:
            self.buckets: List[_Bucket] = []
            return
        # broadcast initial parameters from rank 0
        with torch.no_grad():
            for p in self.params:
                dist.broadcast(p.data, src=0)
        device = self.params[0].device
        cap = int(bucket_cap_mb * 1024 * 1024)
        # reverse order: autograd produces gradients roughly from the last
        # layer backwards, so reverse-order buckets fill (and launch) first
        rev = list(reversed(self.params))
        self.buckets = []
        cur: List[torch.Tensor] = []
        cur_bytes = 0
        for p in rev:
            nbytes = p.numel() * (grad_dtype or p.dtype).itemsize
            if cur and cur_bytes + nbytes > cap:
                self.buckets.append(_Bucket(cur, grad_dtype or cur[0].dtype, device))
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
        if cur:
            self.buckets.append(_Bucket(cur, grad_dtype or cur[0].dtype, device))
        self._param_bucket: Dict[int, tuple] = {}
        for b in self.buckets:
            for i, p in enumerate(b.params):
                self._param_bucket[id(p)] = (b, i)
        self._hooks = []
        for p in self.params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._on_grad_ready)
            )
        self._sync = True
        self._reset_pending()

    def _reset_pending(self):
        for b in self.buckets:
            b.filled = [False] * len(b.params)
            b.work = None
        # collectives must be launched in the SAME order on every rank
        # (RCCL executes them in stream order per communicator) — launch
        # buckets strictly in index order, like torch DDP's sequence rule
        self._next_launch = 0

    @contextlib.contextmanager
    def no_sync(self):
        """Disable per-backward reduction inside gradient-accumulation
        micro-batches: gradients just accumulate into p.grad; the final
        (sync) backward's hooks pick up the accumulated totals. Mirrors
        torch DDP.no_sync() semantics for drivers like run_defect.py with
        --gradient_accumulation_steps > 1."""
        old = self._sync
        self._sync = False
        try:
            yield
        finally:
            self._sync = old

    def _on_grad_ready(self, p: torch.Tensor):
        if not self.enabled or not self._sync:
            return
        b, i = self._param_bucket[id(p)]
        assert not b.filled[i], (
            "gradient produced twice before finalize(); wrap accumulation "
            "micro-batches in engine.no_sync()"
        )
        b.views[i].copy_(p.grad.detach().to(b.flat.dtype))
        b.filled[i] = True
        # fire every consecutively-ready bucket starting at _next_launch
        while self._next_launch < len(self.buckets):
            nb = self.buckets[self._next_launch]
            if not all(nb.filled):
                break
            nb.flat.div_(world_size())
            nb.work = dist.all_reduce(nb.flat, async_op=True)
            self._next_launch += 1

    def finalize(self):
        """Wait for all buckets and write averaged grads back. Call between
        loss.backward() and optimizer.step().

        Every rank MUST call this at each sync boundary even if its backward
        was skipped (e.g. run_defect.py's all-graphs-missing batch skip):
        buckets whose hooks never fired are filled from p.grad (zeros when
        grad is None) and reduced here, so the collectives stay matched
        across ranks."""
        if not self.enabled:
            return
        # launch any buckets the hooks didn't (backward skipped entirely —
        # dummy participation with zero grads — or a frozen/partial subset):
        # fill missing views from p.grad / zeros, in index order
        for b in self.buckets[self._next_launch :]:
            for i, p in enumerate(b.params):
                if not b.filled[i]:
                    if p.grad is None:
                        b.views[i].zero_()
                    else:
                        b.views[i].copy_(p.grad.detach().to(b.flat.dtype))
            b.flat.div_(world_size())
            b.work = dist.all_reduce(b.flat, async_op=True)
        for b in self.buckets:
            b.work.wait()
            for p, v in zip(b.params, b.views):
                if p.grad is None:
                    p.grad = v.to(p.dtype).clone()
                else:
                    p.grad.detach().copy_(v.to(p.grad.dtype))
        self._reset_pending()

    def all_reduce_scalar(self, t: torch.Tensor, op: str = "max") -> torch.Tensor:
        if not self.enabled:
            return t
        dist.all_reduce(t, op=dist.ReduceOp.MAX if op == "max" else dist.ReduceOp.SUM)
        return t
