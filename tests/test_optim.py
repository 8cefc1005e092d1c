"""FlatAdamW vs torch.optim.AdamW/Adam numerics (CPU path here, fused
kernel on GPU in test_gpu_ops-style check)."""

import torch

from deepdfa_amd.parallel.optim import FlatAdamW


def small_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(16, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4)
    )


def run_steps(model, opt, n=5, seed=3):
    gen = torch.Generator().manual_seed(seed)
    for i in range(n):
        x = torch.randn(8, 16, generator=gen)
        loss = model(x).square().mean()
        opt.zero_grad()
        loss.backward()
        opt.step()
    return torch.cat([p.detach().flatten() for p in model.parameters()])


def test_flat_adamw_matches_torch():
    m1, m2 = small_model(), small_model()
    o1 = FlatAdamW(m1.parameters(), lr=1e-2, weight_decay=0.05)
    o2 = torch.optim.AdamW(m2.parameters(), lr=1e-2, weight_decay=0.05)
    p1 = run_steps(m1, o1)
    p2 = run_steps(m2, o2)
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_flat_adam_l2_matches_torch():
    m1, m2 = small_model(1), small_model(1)
    o1 = FlatAdamW(m1.parameters(), lr=1e-3, weight_decay=1e-2, l2_mode=True)
    o2 = torch.optim.Adam(m2.parameters(), lr=1e-3, weight_decay=1e-2)
    p1 = run_steps(m1, o1)
    p2 = run_steps(m2, o2)
    assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_clip_and_state_dict(tmp_path):
    m = small_model(2)
    o = FlatAdamW(m.parameters(), lr=1e-2)
    run_steps(m, o, n=2)
    norm = o.clip_grad_norm_(1e-9)
    assert o.flat_g.abs().max() < 1e-6  # clipped hard
    sd = o.state_dict()
    m2 = small_model(5)
    o2 = FlatAdamW(m2.parameters(), lr=1e-2)
    o2.load_state_dict(sd)
    assert torch.allclose(o2.flat_p, o.flat_p)
    assert o2.step_count == o.step_count
