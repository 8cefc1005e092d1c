"""hipGraph capture of the full flow-GNN train step for Trainer.fit.

The DDFA step is launch-bound (~85 small kernels at batch 256); capturing
fwd+bwd+optimizer into one hipGraph (torch.cuda.CUDAGraph == hipGraph on
ROCm) and replaying removes the per-launch overhead — the measured win in
bench.py round 1 was ~190k -> 196k graphs/s, and this module brings the
same capture to the production fit loop (VERDICT round-1 item 7).

Real training batches vary in (nodes, edges) every step, so batches are
padded to quantized shape buckets (graph/pad.py) and one graph is captured
per bucket; replays copy the padded batch into the bucket's static buffers.
Dummy padding graphs are masked out of loss/metrics exactly
(training_step_masked), so captured training is numerically identical to
eager training on the same batch stream.

Requirements: CUDA device, world_size 1 (no collectives inside the graph),
and a capture-safe optimizer (FlatAdamW: device-side bias correction, flat
zero_grad, sync-free clip).
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from ..graph.batch import BatchedCFG
from ..graph.pad import bucket_shape, pad_batch

__all__ = ["CapturedTrainStep"]

_STATIC_KEYS = ("node_offsets", "indptr", "indices", "t_indptr", "t_indices")


class _Bucket:
    def __init__(self, static_g: BatchedCFG, static_w: torch.Tensor):
        self.static_g = static_g
        self.static_w = static_w
        self.graph: Optional[torch.cuda.CUDAGraph] = None


class CapturedTrainStep:
    """Callable step(batch) that replays a captured hipGraph per shape
    bucket, with eager fallback past max_buckets. Loss is accumulated into
    a device-side scalar inside the graph (read it per-epoch via
    pop_loss_sum() — a per-step host readback would serialize replays)."""

    def __init__(
        self,
        model,
        optimizer,
        batch_size: int,
        grad_clip: Optional[float] = None,
        node_q: int = 1024,
        edge_q: int = 4096,
        max_buckets: int = 16,
        autocast_dtype=torch.bfloat16,
    ):
        assert hasattr(optimizer, "flat_g"), (
            "graph capture needs the capture-safe FlatAdamW "
            "(device-side bias correction)"
        )
        self.model = model
        self.opt = optimizer
        self.b_pad = batch_size + 1  # >= 1 dummy graph absorbs node padding
        self.grad_clip = grad_clip
        self.node_q = node_q
        self.edge_q = edge_q
        self.max_buckets = max_buckets
        self.autocast_dtype = autocast_dtype
        self.device = optimizer.flat_p.device
        self.pool = torch.cuda.graph_pool_handle()
        self.buckets: Dict[Tuple[int, int, int], _Bucket] = {}
        self.loss_accum = torch.zeros((), dtype=torch.float32, device=self.device)
        self.steps = 0
        self.eager_steps = 0

    # -- public ---------------------------------------------------------------

    def __call__(self, batch) -> None:
        g, extra = batch if isinstance(batch, tuple) else (batch, {})
        if g.num_graphs >= self.b_pad:
            self._eager(g, extra)
            return
        shape = bucket_shape(g, self.b_pad, self.node_q, self.edge_q)
        bucket = self.buckets.get(shape)
        padded, w = pad_batch(g, *shape)
        if bucket is None:
            if len(self.buckets) >= self.max_buckets:
                self._eager(g, extra)
                return
            bucket = self._capture(shape, padded, w)
        else:
            self._fill(bucket, padded, w)
            bucket.graph.replay()
        self.steps += 1

    def pop_loss_sum(self) -> float:
        """Host-read and reset the in-graph loss accumulator (syncs once)."""
        v = float(self.loss_accum.item())
        self.loss_accum.zero_()
        return v

    # -- internals ------------------------------------------------------------

    def _eager(self, g, extra) -> None:
        with torch.autocast(device_type="cuda", dtype=self.autocast_dtype):
            loss = self.model.training_step((g.to(self.device), extra))
        self.opt.zero_grad()
        loss.backward()
        if self.grad_clip:
            self.opt.clip_grad_norm_(self.grad_clip)
        self.opt.step()
        self.loss_accum += loss.detach().float()
        self.steps += 1
        self.eager_steps += 1

    def _fill(self, bucket: _Bucket, padded: BatchedCFG, w: torch.Tensor) -> None:
        sg = bucket.static_g
        for k in _STATIC_KEYS:
            getattr(sg, k).copy_(getattr(padded, k), non_blocking=True)
        for k, v in padded.ndata.items():
            sg.ndata[k].copy_(v, non_blocking=True)
        bucket.static_w.copy_(w, non_blocking=True)

    def _capture(self, shape, padded: BatchedCFG, w: torch.Tensor) -> _Bucket:
        static_g = padded.to(self.device)
        static_w = w.to(self.device)
        bucket = _Bucket(static_g, static_w)
        self.buckets[shape] = bucket

        from ..ops import transformer as _tops

        _tops.CAPTURE_REFRESH[0] = True  # weight casts re-run inside the graph

        def run():
            with torch.autocast(device_type="cuda", dtype=self.autocast_dtype):
                loss = self.model.training_step_masked(static_g, {}, static_w)
            self.opt.zero_grad()
            loss.backward()
            if self.grad_clip:
                self.opt.clip_grad_norm_(self.grad_clip)
            self.opt.step()
            self.loss_accum += loss.detach().float()

        # warm up the exact captured sequence on a side stream (allocator +
        # autotune caches settle), then capture; the warmup mutated real
        # state — that's fine, it is a legitimate training step
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            run()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        cg = torch.cuda.CUDAGraph()
        with torch.cuda.graph(cg, pool=self.pool):
            run()
        bucket.graph = cg
        # the warmup run WAS this batch's real training step (stream capture
        # records without executing), so the caller must not replay again
        return bucket
