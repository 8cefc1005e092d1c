"""Within-probe A/B of flash_dkv variants (guide §5.4 rule 24).

Variants (DFA_DKV_VARIANT): 0 = direct L1 scalar dob/qb loads,
2 = LDS-bounce vector B-fragment reads, 3 = 0 + s_setprio on younger waves.
Also numerics-checks each variant against variant 0.
Run on a GPU box: python tools/dkv_probe.py
"""

import os
import sys
import time

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

VARS = ["4", "6"]


def run_bwd():
    return ext.flash_attn_bwd(dO, q, k, v, O, lse, H, valid, None, 0.125,
                              False, 0.0, 0, False, False)


# numerics: each variant vs variant 0
os.environ["DFA_DKV_VARIANT"] = "0"
ref = run_bwd()
for vr in VARS[1:]:
    os.environ["DFA_DKV_VARIANT"] = vr
    out = run_bwd()
    for name, a, b in (("dq", ref[0], out[0]), ("dk", ref[1], out[1]),
                       ("dv", ref[2], out[2])):
        if not torch.equal(a, b):
            md = (a.float() - b.float()).abs().max().item()
            print(f"variant {vr} {name}: MISMATCH max|diff|={md:.3e}")
            if md > 1e-3:
                sys.exit(1)
print("numerics ok (bitwise vs variant 0 unless noted)")

# timing: interleaved rounds, median
for _ in range(3):
    for vr in VARS:
        os.environ["DFA_DKV_VARIANT"] = vr
        run_bwd()
torch.cuda.synchronize()
results = {vr: [] for vr in VARS}
for rnd in range(7):
    for vr in VARS:
        os.environ["DFA_DKV_VARIANT"] = vr
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(10):
            run_bwd()
        torch.cuda.synchronize()
        results[vr].append((time.perf_counter() - t0) / 10 * 1e6)
print(f"{'variant':<10}{'median_us(bwd total)':>22}{'min_us':>9}")
for vr, ts in results.items():
    ts.sort()
    print(f"{vr:<10}{ts[len(ts)//2]:>22.1f}{ts[0]:>9.1f}")

# segment breakdown (variant 9, instrumented): cycles per segment
os.environ["DFA_DKV_VARIANT"] = "9"
run_bwd()
torch.cuda.synchronize()
ext.dkv_prof()  # discard warmup
for _ in range(10):
    run_bwd()
torch.cuda.synchronize()
seg = ext.dkv_prof().tolist()
names = ["frag+lse loads", "P recompute", "dP MFMAs", "pd/ds+LDS writes",
         "pa/dsa+dob/qb loads", "dV/dK MFMAs", "reduce+store epilogue",
         "loop head"]
tot = sum(seg)
print("\nvariant-9 segment breakdown (10 calls):")
for n, s in sorted(zip(names, seg), key=lambda x: -x[1]):
    print(f"  {n:<24}{s:>14}  {100.0*s/max(1,tot):5.1f}%")
os.environ["DFA_DKV_VARIANT"] = "9"
torch.cuda.synchronize()
t0 = time.perf_counter()
for _ in range(10):
    run_bwd()
torch.cuda.synchronize()
print(f"variant 9 wall: {(time.perf_counter()-t0)/10*1e6:.1f} us (incl. overhead)")
