"""Flat fused AdamW/Adam (MI355X-native optimizer).

All parameters are re-viewed into ONE contiguous fp32 master buffer at
construction (param .data becomes a view; .grad is pre-assigned a view of a
flat gradient buffer, which autograd accumulates into). The step is then a
single fused HIP kernel (csrc/adamw.hip), zero_grad is one fill, the global
grad norm is one reduction, and data-parallel gradient averaging is ONE
flat RCCL all-reduce (`allreduce_grads`).

288 GB HBM per GPU makes the flat fp32 layout free even for the largest
model here (~224M params => ~3.6 GB of param+grad+state).
"""

from __future__ import annotations

from typing import Iterable, Optional

import torch


class FlatAdamW:
    def __init__(
        self,
        params: Iterable[torch.nn.Parameter],
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        l2_mode: bool = False,  # True = classic Adam (L2 in gradient)
    ):
        self.params = [p for p in params if p.requires_grad]
        assert self.params, "no trainable parameters"
        device = self.params[0].device
        total = sum(p.numel() for p in self.params)
        self.flat_p = torch.empty(total, dtype=torch.float32, device=device)
        self.flat_g = torch.zeros(total, dtype=torch.float32, device=device)
        self.m = torch.zeros(total, dtype=torch.float32, device=device)
        self.v = torch.zeros(total, dtype=torch.float32, device=device)
        off = 0
        for p in self.params:
            n = p.numel()
            self.flat_p[off : off + n].copy_(p.data.float().flatten())
            p.data = self.flat_p[off : off + n].view(p.shape)
            p.grad = self.flat_g[off : off + n].view(p.shape)
            off += n
        # bf16 shadow of the flat master params, written by the adamw kernel
        # itself each step (+2 B/param on a 32 B/param pass): the fused-linear
        # weight views (ops/transformer.py) read from it, replacing per-weight
        # fp32->bf16 cast kernels every step. `_dfa_w16_ver` records the param
        # version the shadow was last synced at so torch-level writes
        # (checkpoint loads) are detected and re-cast by the consumer.
        self.flat_p16 = None
        if device.type == "cuda":
            self.flat_p16 = self.flat_p.to(torch.bfloat16)
            off = 0
            for p in self.params:
                n = p.numel()
                p._dfa_w16 = self.flat_p16[off : off + n].view(p.shape)
                p._dfa_w16_ver = p._version
                p._dfa_w16_base = self.flat_p16
                p._dfa_w16_off = off
                # flat-grad view bookkeeping for DIRECT gradient accumulation:
                # weight-grad kernels atomically add into .grad (pre-zeroed by
                # zero_grad's single fill) and return None to autograd,
                # skipping the per-param zeros() + AccumulateGrad add pair
                p._dfa_gbase = self.flat_g
                p._dfa_goff = off
                off += n
        self.lr = lr
        self.betas = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.l2_mode = l2_mode
        self.step_count = 0
        self.param_groups = [{"params": self.params, "lr": lr}]  # LR-sched compat

    def zero_grad(self, set_to_none: bool = False):
        self.flat_g.zero_()

    def grad_norm(self) -> torch.Tensor:
        return torch.linalg.vector_norm(self.flat_g)

    def clip_grad_norm_(self, max_norm: float) -> torch.Tensor:
        # capture-safe: no host readback — clamp the coefficient on-device.
        # On GPU the scaling is FOLDED into the adamw kernel's gradient
        # load (a separate mul_ pass re-read+wrote the whole flat gradient:
        # 1.8 GB at 223M params); the CPU path scales in place.
        norm = self.grad_norm()
        coef = torch.clamp(max_norm / (norm + 1e-6), max=1.0)
        if self.flat_p.is_cuda:
            self._clip_coef = coef.reshape(1)
        else:
            self.flat_g.mul_(coef)
        return norm

    def allreduce_grads(self):
        import torch.distributed as dist

        if dist.is_available() and dist.is_initialized() and dist.get_world_size() > 1:
            self.flat_g.div_(dist.get_world_size())
            dist.all_reduce(self.flat_g)

    @torch.no_grad()
    def step(self, closure=None):
        self.step_count += 1
        lr = self.param_groups[0]["lr"]
        b1, b2 = self.betas
        if self.flat_p.is_cuda:
            from ..ops import load_ext
            from ..ops.transformer import bump_weights_epoch

            ext = load_ext(required=True)
            # bias corrections computed IN the adamw kernel from a device
            # step counter: correct under hipGraph replay (the captured
            # add_ advances every replay; a python int would freeze at
            # capture time), and only ONE tiny elementwise node remains
            if not hasattr(self, "_step_t") or not self._step_t.is_cuda:
                self._step_t = torch.zeros(1, device=self.flat_p.device)
                self._step_t.fill_(float(self.step_count - 1))
            self._step_t.add_(1.0)
            gclip = getattr(self, "_clip_coef", None)
            ext.adamw_fused(
                self.flat_p, self.flat_g, self.m, self.v, lr, b1, b2, self.eps,
                self.weight_decay, self._step_t, gclip, self.l2_mode,
                self.flat_p16,
            )
            self._clip_coef = None
            # the raw kernel write bypasses torch version counters —
            # invalidate the bf16 weight-cast caches explicitly
            bump_weights_epoch()
            return
        # CPU reference path (same math)
        g = self.flat_g
        if self.l2_mode:
            g = g + self.weight_decay * self.flat_p
        self.m.mul_(b1).add_(g, alpha=1 - b1)
        self.v.mul_(b2).addcmul_(g, g, value=1 - b2)
        bc1 = 1 - b1 ** self.step_count
        bc2 = 1 - b2 ** self.step_count
        if not self.l2_mode:
            self.flat_p.mul_(1 - lr * self.weight_decay)
        denom = (self.v / bc2).sqrt_().add_(self.eps)
        self.flat_p.addcdiv_(self.m / bc1, denom, value=-lr)

    # -- minimal state-dict compat -------------------------------------------

    def state_dict(self):
        # clone: the live buffers keep mutating after a checkpoint snapshot
        return {
            "flat_p": self.flat_p.detach().clone(),
            "m": self.m.detach().clone(),
            "v": self.v.detach().clone(),
            "step_count": self.step_count,
            "lr": self.param_groups[0]["lr"],
        }

    def load_state_dict(self, sd):
        self.flat_p.copy_(sd["flat_p"])
        self.m.copy_(sd["m"])
        self.v.copy_(sd["v"])
        self.step_count = sd["step_count"]
        self.param_groups[0]["lr"] = sd["lr"]
        # the device-side step counter drives GPU bias correction — resync
        # it so a resume doesn't step with a stale count
        if hasattr(self, "_step_t"):
            del self._step_t
