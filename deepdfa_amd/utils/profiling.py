"""FLOPs/latency profiling (reference §5.1 parity).

The reference attaches DeepSpeed's FlopsProfiler (base_module.py:76-77,
linevul_main.py:332-394) and CUDA-event timing, appending JSON lines to
profiledata.jsonl / timedata.jsonl which scripts/report_profiling.py
aggregates to GFLOPs and ms/example. Same on-disk contract here, with a
module-hook MAC counter (Linear/Embedding + attention score/context
matmuls) and hipEvent timing (torch.cuda.Event on ROCm).
"""

from __future__ import annotations

import json
import time
from typing import Optional

import torch
from torch import nn


class FlopsProfiler:
    """Counts MACs of one forward via module hooks."""

    def __init__(self, model: nn.Module):
        self.model = model
        self.macs = 0
        self.params = sum(p.numel() for p in model.parameters())
        self._hooks = []

    def start_profile(self):
        self.macs = 0
        self._hooks = []

        def linear_hook(mod, inp, out):
            self.macs += inp[0].numel() // inp[0].shape[-1] * mod.in_features * mod.out_features

        def embedding_hook(mod, inp, out):
            self.macs += out.numel()  # gather cost proxy (matches deepspeed)

        def gru_hook(mod, inp, out):
            n = inp[0].shape[0]
            self.macs += n * 6 * mod.hidden_size * mod.hidden_size

        for m in self.model.modules():
            if isinstance(m, nn.Linear):
                self._hooks.append(m.register_forward_hook(linear_hook))
            elif isinstance(m, nn.Embedding):
                self._hooks.append(m.register_forward_hook(embedding_hook))
            elif isinstance(m, nn.GRUCell):
                self._hooks.append(m.register_forward_hook(gru_hook))
        # attention score/context matmuls (B*H*Lq*Lk*d each)
        try:
            from ..models.roberta import RobertaSelfAttention

            def attn_hook(mod, inp, out):
                x = inp[0]
                B, L, D = x.shape
                self.macs += 2 * B * mod.num_heads * L * L * mod.head_dim

            for m in self.model.modules():
                if isinstance(m, RobertaSelfAttention):
                    self._hooks.append(m.register_forward_hook(attn_hook))
        except ImportError:
            pass

    def stop_profile(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []

    def get_total_flops(self):
        return 2 * self.macs

    def get_total_macs(self):
        return self.macs

    def get_total_params(self):
        return self.params

    def end_profile(self):
        self.stop_profile()


class ProfilingWriter:
    """Appends profile/time JSON lines (reference jsonl contract)."""

    def __init__(self, profile_path="profiledata.jsonl", time_path="timedata.jsonl"):
        self.profile_path = profile_path
        self.time_path = time_path

    def write_profile(self, flops, macs, params, batch_size, time_ms: Optional[float] = None):
        with open(self.profile_path, "a") as f:
            f.write(
                json.dumps(
                    {
                        "flops": flops,
                        "macs": macs,
                        "params": params,
                        "batch_size": int(batch_size),
                        "time_ms": time_ms,
                    }
                )
                + "\n"
            )

    def write_time(self, batch_size, time_ms):
        with open(self.time_path, "a") as f:
            f.write(json.dumps({"batch_size": int(batch_size), "time_ms": time_ms}) + "\n")


class CudaTimer:
    """hipEvent bracket (falls back to wall time on CPU)."""

    def __init__(self):
        self.use_cuda = torch.cuda.is_available()
        if self.use_cuda:
            self.start_ev = torch.cuda.Event(enable_timing=True)
            self.end_ev = torch.cuda.Event(enable_timing=True)

    def __enter__(self):
        if self.use_cuda:
            self.start_ev.record()
        else:
            self.t0 = time.perf_counter()
        return self

    def __exit__(self, *a):
        if self.use_cuda:
            self.end_ev.record()
            torch.cuda.synchronize()
            self.ms = self.start_ev.elapsed_time(self.end_ev)
        else:
            self.ms = (time.perf_counter() - self.t0) * 1000.0
