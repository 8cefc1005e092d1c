"""Flash attention numerics vs the materialized-P oracle (GPU)."""

import math

import pytest
import torch

pytestmark = pytest.mark.gpu

bf = torch.bfloat16


@pytest.fixture(scope="module")
def dev():
    assert torch.cuda.is_available()
    return torch.device("cuda:0")


def materialized(q, k, v, H, valid, bias, scale, causal):
    """fp32 oracle in (B, L, H*d) layout."""
    B, L, HD = q.shape
    d = HD // H

    def split(t):
        return t.float().view(B, L, H, d).transpose(1, 2)

    qs, ks, vs = split(q), split(k), split(v)
    s = torch.matmul(qs, ks.transpose(-1, -2)) * scale
    if bias is not None:
        s = s + bias.unsqueeze(0)
    if valid is not None:
        mask = torch.arange(L, device=q.device).view(1, 1, 1, L) >= valid.view(-1, 1, 1, 1)
        s = s.masked_fill(mask, float("-inf"))
    if causal:
        cm = torch.arange(L, device=q.device).view(1, L) > torch.arange(
            L, device=q.device
        ).view(L, 1)
        s = s.masked_fill(cm.view(1, 1, L, L), float("-inf"))
    p = torch.nan_to_num(torch.softmax(s, -1), nan=0.0)
    o = torch.matmul(p, vs)
    return o.transpose(1, 2).reshape(B, L, HD)


def rand_qkv(dev, B=2, H=4, L=128, seed=0, grad=False):
    torch.manual_seed(seed)
    mk = lambda: (torch.randn(B, L, H * 64, device=dev) * 0.5).to(bf).requires_grad_(grad)  # noqa: E731
    return mk(), mk(), mk()


@pytest.mark.parametrize("causal", [False, True])
def test_flash_fwd_parity(dev, causal):
    from deepdfa_amd.ops.transformer import flash_attention

    q, k, v = rand_qkv(dev, seed=1)
    valid = torch.tensor([128, 70], dtype=torch.int32, device=dev)
    scale = 1.0 / math.sqrt(64)
    out = flash_attention(q, k, v, 4, valid=valid, scale=scale, causal=causal)
    ref_o = materialized(q, k, v, 4, valid, None, scale, causal)
    # rows beyond valid are undefined-but-finite; compare valid rows
    m = (torch.arange(128, device=dev).view(1, -1, 1) < valid.view(-1, 1, 1)).float()
    err = ((out.float() - ref_o) * m).abs().max().item()
    assert err < 3e-2, err


def test_flash_fwd_bias(dev):
    from deepdfa_amd.ops.transformer import flash_attention

    q, k, v = rand_qkv(dev, seed=2)
    bias = torch.randn(4, 128, 128, device=dev)
    out = flash_attention(q, k, v, 4, bias=bias.transpose(-1, -2).contiguous(),
                          scale=1.0)
    ref_o = materialized(q, k, v, 4, None, bias, 1.0, False)
    err = (out.float() - ref_o).abs().max().item()
    assert err < 5e-2, err


def test_flash_bwd_parity(dev):
    from deepdfa_amd.ops.transformer import flash_attention

    scale = 1.0 / math.sqrt(64)
    valid = torch.tensor([128, 100], dtype=torch.int32, device=dev)
    q, k, v = rand_qkv(dev, seed=3, grad=True)
    out = flash_attention(q, k, v, 4, valid=valid, scale=scale)
    # mask pad-query-row grads on BOTH sides: those rows' outputs are
    # downstream-masked in the models, so their grads are not defined parity
    m = (torch.arange(128, device=dev).view(1, -1, 1) < valid.view(-1, 1, 1)).float()
    go = torch.randn_like(out)
    gom = (go.float() * m).to(bf)
    out.backward(gom)
    q2 = q.detach().clone().requires_grad_(True)
    k2 = k.detach().clone().requires_grad_(True)
    v2 = v.detach().clone().requires_grad_(True)
    ref_o = materialized(q2, k2, v2, 4, valid, None, scale, False)
    ref_o.backward(gom.float())
    for a, b, name in ((q, q2, "dq"), (k, k2, "dk"), (v, v2, "dv")):
        err = (a.grad.float() - b.grad).abs().max().item()
        ref_mag = b.grad.abs().max().item()
        assert err < 0.05 * max(ref_mag, 1.0), (name, err, ref_mag)


def test_flash_bwd_bias_grad(dev):
    from deepdfa_amd.ops.transformer import flash_attention

    q, k, v = rand_qkv(dev, B=2, seed=4)
    bias = torch.randn(4, 128, 128, device=dev, requires_grad=True)
    out = flash_attention(q, k, v, 4, bias=bias.transpose(-1, -2).contiguous(),
                          scale=1.0, causal=True)
    go = torch.randn_like(out)
    out.backward(go)
    b2 = bias.detach().clone().requires_grad_(True)
    ref_o = materialized(q, k, v, 4, None, b2, 1.0, True)
    ref_o.backward(go.float())
    err = (bias.grad - b2.grad).abs().max().item()
    assert err < 0.05 * max(b2.grad.abs().max().item(), 1.0), err


def test_flash_dropout_consistency(dev):
    from deepdfa_amd.ops.transformer import flash_attention

    q, k, v = rand_qkv(dev, seed=5, grad=True)
    out = flash_attention(q, k, v, 4, scale=0.125, dropout_p=0.5)
    out_ref = flash_attention(q.detach(), k.detach(), v.detach(), 4, scale=0.125)
    # dropout changes the output meaningfully but keeps it finite
    assert torch.isfinite(out.float()).all()
    assert (out.float() - out_ref.float()).abs().max().item() > 0.01
    out.float().square().mean().backward()
    assert torch.isfinite(q.grad.float()).all()


def test_roberta_flash_vs_materialized(dev):
    """Whole encoder: flash path (L=128) vs materialized (output_attentions)."""
    from deepdfa_amd.models.roberta import RobertaConfig, RobertaModel, init_roberta_weights

    torch.manual_seed(0)
    cfg = RobertaConfig(vocab_size=500, hidden_size=256, num_hidden_layers=2,
                        num_attention_heads=4, intermediate_size=512,
                        max_position_embeddings=200)
    model = RobertaModel(cfg)
    init_roberta_weights(model)
    model = model.to(dev).eval()
    ids = torch.randint(3, 500, (2, 128), device=dev)
    ids[0, 90:] = 1
    with torch.no_grad(), torch.autocast(device_type="cuda", dtype=bf):
        out_flash, _ = model(ids)  # flash (no attention output)
        out_mat, _ = model(ids, output_attentions=True)  # materialized
    mask = ids.ne(1).unsqueeze(-1)
    err = ((out_flash.float() - out_mat.float()) * mask).abs().max().item()
    assert err < 0.05, err


@pytest.mark.gpu
def test_flash_strided_qkv_views_match_contiguous():
    """Row-strided Q/K/V (slices of a fused QKV buffer) must give the same
    forward and backward as contiguous tensors."""
    from deepdfa_amd.ops.transformer import flash_attention

    torch.manual_seed(0)
    B, L, H, d = 2, 128, 4, 64
    D = H * d
    qkv = (torch.randn(B, L, 3 * D, device="cuda", dtype=torch.bfloat16) * 0.3
           ).requires_grad_()
    qc = qkv[..., :D].detach().contiguous().requires_grad_()
    kc = qkv[..., D:2 * D].detach().contiguous().requires_grad_()
    vc = qkv[..., 2 * D:].detach().contiguous().requires_grad_()
    valid = torch.tensor([L, 96], dtype=torch.int32, device="cuda")

    o1 = flash_attention(qkv[..., :D], qkv[..., D:2 * D], qkv[..., 2 * D:],
                         H, valid=valid, scale=0.125)
    o2 = flash_attention(qc, kc, vc, H, valid=valid, scale=0.125)
    assert torch.equal(o1, o2)
    g = torch.randn_like(o1)
    o1.backward(g)
    o2.backward(g)
    dq = qkv.grad[..., :D]
    dk = qkv.grad[..., D:2 * D]
    dv = qkv.grad[..., 2 * D:]
    assert torch.equal(dq, qc.grad)
    assert torch.equal(dk, kc.grad)
    assert torch.equal(dv, vc.grad)
