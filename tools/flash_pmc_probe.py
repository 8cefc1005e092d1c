"""Run ONLY the flash-attention bwd kernels repeatedly for a PMC capture.

Usage (on GPU box):
  cd /tmp && export TMPDIR=/tmp
  rocprofv3 --pmc SQ_WAVE_CYCLES SQ_WAIT_ANY SQ_WAIT_INST_ANY \
      SQ_ACTIVE_INST_ANY SQ_VALU_MFMA_BUSY_CYCLES SQ_LDS_BANK_CONFLICT \
      -d /tmp/fpmc -- python /root/repo/tools/flash_pmc_probe.py
  python /root/repo/tools/pmc_summarize.py /tmp/fpmc
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from deepdfa_amd.ops import load_ext

ext = load_ext(required=True)
dev = "cuda"
bf = torch.bfloat16
torch.manual_seed(0)
B, L, H, d = 16, 512, 12, 64

q = torch.randn(B, L, H * d, device=dev, dtype=bf) * 0.3
k = torch.randn(B, L, H * d, device=dev, dtype=bf) * 0.3
v = torch.randn(B, L, H * d, device=dev, dtype=bf) * 0.3
dO = torch.randn(B, L, H * d, device=dev, dtype=bf) * 0.3
valid = torch.full((B,), L, dtype=torch.int32, device=dev)
O, lse = ext.flash_attn_fwd(q, k, v, H, valid, None, 0.125, False, 0.0, 0)
torch.cuda.synchronize()
which = os.environ.get("FLASH_PMC_WHICH", "bwd")
for _ in range(30):
    if which == "fwd":
        ext.flash_attn_fwd(q, k, v, H, valid, None, 0.125, False, 0.0, 0)
    else:
        ext.flash_attn_bwd(dO, q, k, v, O, lse, H, valid, None, 0.125, False, 0.0, 0, False, False)
torch.cuda.synchronize()
print("done")
