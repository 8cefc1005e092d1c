"""CPU tests for the bf16-shadow / cast-cache invalidation protocol
(ops/transformer.py _shadow_w16/_bias_f32, models/t5.py _flash_bias_T):
the version-counter bookkeeping must detect torch-level writes and the
per-step caches must reuse without copies otherwise."""

import torch

from deepdfa_amd.models.t5 import _flash_bias_T
from deepdfa_amd.ops.transformer import _bias_f32, _shadow_w16


def _attach_shadow(p):
    """Simulate what FlatAdamW does on CUDA (parallel/optim.py)."""
    flat16 = p.detach().to(torch.bfloat16).flatten().clone()
    p._dfa_w16 = flat16.view(p.shape)
    p._dfa_w16_ver = p._version
    p._dfa_w16_base = flat16
    p._dfa_w16_off = 0
    return p


def test_shadow_resyncs_after_external_write():
    p = _attach_shadow(torch.nn.Parameter(torch.randn(8, 8)))
    w16 = _shadow_w16(p)
    assert torch.equal(w16.float(), p.detach().to(torch.bfloat16).float())
    # torch-level write (checkpoint load) bumps the version counter
    with torch.no_grad():
        p.copy_(torch.randn(8, 8))
    w16b = _shadow_w16(p)
    assert w16b is w16  # same buffer, refreshed in place
    assert torch.equal(w16b.float(), p.detach().to(torch.bfloat16).float())


def test_shadow_not_recopied_when_clean():
    p = _attach_shadow(torch.nn.Parameter(torch.randn(4, 4)))
    _shadow_w16(p)
    ver = p._dfa_w16_ver
    # poison the shadow; without a version bump it must NOT be re-synced
    # (the optimizer kernel, not torch, owns shadow freshness)
    p._dfa_w16.fill_(0)
    w16 = _shadow_w16(p)
    assert p._dfa_w16_ver == ver and float(w16.float().abs().sum()) == 0.0


def test_shadow_none_without_flat_optimizer():
    p = torch.nn.Parameter(torch.randn(4, 4))
    assert _shadow_w16(p) is None


def test_bias_f32_no_copy_for_master_fp32():
    b = torch.nn.Parameter(torch.randn(16))
    out = _bias_f32(b)
    assert out.data_ptr() == b.data_ptr()  # in-place view, no copy
    bf = torch.nn.Parameter(torch.randn(16).to(torch.bfloat16))
    out2 = _bias_f32(bf)
    assert out2.dtype == torch.float32 and out2.data_ptr() != bf.data_ptr()
    assert _bias_f32(None) is None


def test_flash_bias_transpose_cached_per_tensor():
    pb = torch.randn(1, 4, 8, 8)
    accum = torch.zeros(4, 8, 8)
    bT = _flash_bias_T(pb, accum)
    assert bT.shape == (4, 8, 8)
    assert torch.equal(bT, pb.detach().squeeze(0).transpose(-1, -2))
    assert _flash_bias_T(pb, accum) is bT  # cached on the tensor
    pb2 = torch.randn(1, 4, 8, 8)
    assert _flash_bias_T(pb2, accum) is not bT  # fresh per-step tensor
    # with no accum (differentiable path) the transpose stays in the graph
    pb3 = torch.randn(1, 4, 8, 8, requires_grad=True)
    bT3 = _flash_bias_T(pb3, None)
    bT3.square().sum().backward()
    assert pb3.grad is not None and pb3.grad.abs().sum() > 0


def test_flat_adamw_cpu_has_no_shadow():
    from deepdfa_amd.parallel.optim import FlatAdamW

    lin = torch.nn.Linear(8, 8)
    opt = FlatAdamW(lin.parameters(), lr=1e-3)
    assert opt.flat_p16 is None
    assert not hasattr(lin.weight, "_dfa_w16")
    # CPU step still works (reference-path math)
    lin(torch.randn(4, 8)).sum().backward()
    opt.step()
