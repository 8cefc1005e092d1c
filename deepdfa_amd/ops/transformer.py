"""Autograd ops for the transformer stacks.

GPU: hand-written HIP kernels (csrc/transformer_kernels.hip), bf16/fp32 IO
with fp32 statistics and fp32 LN/bias parameters (mixed-precision master
weights). CPU: plain torch (also the numerics oracle).
"""

from __future__ import annotations

from typing import Optional

import torch

from ._ext import load_ext


class _LayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        ext = load_ext(required=True)
        wf = weight.float().contiguous()
        bf = bias.float().contiguous()
        x = x.contiguous()
        y, mean, rstd = ext.layernorm_fwd(x, wf, bf, eps)
        ctx.save_for_backward(x, wf, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        x, wf, mean, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        dx = ext.layernorm_bwd(dy, x, wf, mean, rstd)
        dgamma, dbeta = ext.layernorm_wgrad(dy, x, mean, rstd)
        return dx, dgamma, dbeta, None


def layer_norm(x, weight, bias, eps: float = 1e-5):
    if x.is_cuda:
        return _LayerNorm.apply(x, weight, bias, eps)
    return torch.nn.functional.layer_norm(
        x.float(), (x.shape[-1],), weight.float(), bias.float(), eps
    ).to(x.dtype)


class _BiasGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        ext = load_ext(required=True)
        bf = bias.float().contiguous()
        x = x.contiguous()
        ctx.save_for_backward(x, bf)
        return ext.bias_gelu_fwd(x, bf)

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        x, bf = ctx.saved_tensors
        dy = dy.contiguous()
        dx = ext.bias_gelu_bwd(dy, x, bf)
        dbias = ext.colsum(dx.view(-1, dx.shape[-1]))
        return dx, dbias


def bias_gelu(x, bias):
    if x.is_cuda:
        return _BiasGelu.apply(x, bias)
    return torch.nn.functional.gelu(x.float() + bias.float()).to(x.dtype)


# cast-cache invalidation epoch: optimizers that write master weights
# through raw extension kernels (FlatAdamW) bypass torch's version
# counters, so they bump this instead
_weights_epoch = [0]

# hipGraph capture mode: when True, fused_linear re-casts weights INTO its
# cached buffers every call so the cast is captured and replays against the
# updated master weights
CAPTURE_REFRESH = [False]


def bump_weights_epoch():
    _weights_epoch[0] += 1


def _shadow_w16(weight):
    """The optimizer-maintained flat bf16 shadow view of `weight`
    (parallel/optim.py FlatAdamW), re-synced if a torch-level write
    (checkpoint load, broadcast) moved the parameter's version counter
    since the last sync. None when no flat optimizer owns the weight."""
    w16 = getattr(weight, "_dfa_w16", None)
    if w16 is None:
        return None
    if weight._version != weight._dfa_w16_ver:
        w16.copy_(weight.detach())
        weight._dfa_w16_ver = weight._version
    return w16


def _bias_f32(bias):
    """Master-fp32 biases are used in place (no copy); anything else is
    cast-copied (rare: only models trained without the flat optimizer)."""
    if bias is None:
        return None
    b = bias.detach()
    if b.dtype == torch.float32 and b.is_contiguous():
        return b
    return b.float().contiguous()


_seed_state = {"base": None, "ctr": 0}


def _next_seed() -> int:
    if _seed_state["base"] is None:
        _seed_state["base"] = torch.initial_seed() & 0x7FFFFFFF
    _seed_state["ctr"] += 1
    return (_seed_state["base"] * 2654435761 + _seed_state["ctr"]) & 0x7FFFFFFFFFFFFFFF


class _MaskedSoftmaxDropout(torch.autograd.Function):
    """(P, Pd) = softmax(S * scale) with key positions >= valid[b] masked,
    plus fused attention dropout (mask regenerated from the seed in
    backward; no mask storage). Pd is the differentiable output feeding
    P @ V; P is the pre-dropout probabilities (analysis output, marked
    non-differentiable)."""

    @staticmethod
    def forward(ctx, S, valid, scale, dropout_p, causal=False):
        ext = load_ext(required=True)
        seed = _next_seed() if dropout_p > 0 else 0
        P, Pd = ext.softmax_mask_fwd(S.contiguous(), valid, scale, dropout_p, seed, causal)
        ctx.save_for_backward(P)
        ctx.scale = scale
        ctx.dropout_p = dropout_p
        ctx.seed = seed
        ctx.mark_non_differentiable(P)
        return P, Pd

    @staticmethod
    def backward(ctx, _dP_unused, dPd):
        ext = load_ext(required=True)
        (P,) = ctx.saved_tensors
        dS = ext.softmax_mask_bwd(dPd.contiguous(), P, ctx.scale, ctx.dropout_p, ctx.seed)
        return dS, None, None, None, None


def masked_softmax_dropout(
    S: torch.Tensor,
    valid: Optional[torch.Tensor],
    scale: float,
    dropout_p: float = 0.0,
    causal: bool = False,
):
    """Returns (P pre-dropout, Pd post-dropout)."""
    if S.is_cuda:
        return _MaskedSoftmaxDropout.apply(S, valid, scale, dropout_p, causal)
    s = S.float() * scale
    L = S.shape[-1]
    if valid is not None:
        mask = torch.arange(L).view(1, 1, 1, L) >= valid.view(-1, 1, 1, 1)
        s = s.masked_fill(mask, float("-inf"))
    if causal:
        Lq = S.shape[-2]
        cmask = torch.arange(L).view(1, L) > torch.arange(Lq).view(Lq, 1)
        s = s.masked_fill(cmask.view(1, 1, Lq, L), float("-inf"))
    p = torch.softmax(s, dim=-1)
    p = torch.nan_to_num(p, nan=0.0).to(S.dtype)
    pd = torch.nn.functional.dropout(p, dropout_p) if dropout_p > 0 else p
    return p, pd


class _LinearBf16(torch.autograd.Function):
    """Linear layer on the custom transformer-shape GEMMs: forward and
    input-grad via gemm2 (2-phase glds MFMA), weight grad via the split-K
    wgrad kernel (fp32 output — the master-weight grad dtype directly),
    bias grad via colsum."""

    @staticmethod
    def forward(ctx, x, weight, bias):
        ext = load_ext(required=True)
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        # fat shapes route to the library GEMMs and never read wt16 (backward
        # uses w16 directly), so the transpose copy is skipped for them
        fat = weight.shape[0] > 1024 or weight.shape[1] > 1024
        w16 = _shadow_w16(weight)
        if w16 is not None:
            # optimizer-maintained bf16 shadow (parallel/optim.py): no cast
            # kernel — the adamw kernel wrote this view at the last step.
            # Only the transposed copy (square shapes) needs per-step refresh.
            b32 = _bias_f32(bias)
            if fat:
                wt16 = None
            else:
                cache = getattr(weight, "_dfa_wt_cache", None)
                key = (weight._version, _weights_epoch[0])
                if cache is None:
                    wt16 = w16.t().contiguous()
                    weight._dfa_wt_cache = (key, wt16)
                elif CAPTURE_REFRESH[0] or cache[0] != key:
                    _, wt16 = cache
                    wt16.copy_(w16.t())
                    weight._dfa_wt_cache = (key, wt16)
                else:
                    _, wt16 = cache
        else:
            # no flat optimizer: cache the bf16 / transposed weight per
            # parameter VERSION — the cast and transpose copies otherwise
            # dominate the elementwise kernel count (~3 x 72 linears/step)
            cache = getattr(weight, "_dfa_cast_cache", None)
            key = (weight._version, _weights_epoch[0])
            if cache is None:
                w16 = weight.detach().to(torch.bfloat16).contiguous()
                wt16 = None if fat else w16.t().contiguous()
                b32 = bias.detach().float().contiguous() if bias is not None else None
                weight._dfa_cast_cache = (key, w16, wt16, b32)
            elif CAPTURE_REFRESH[0] or cache[0] != key:
                # refresh INTO the cached buffers: under hipGraph capture the
                # cast must be part of the graph (stable addresses, re-run
                # each replay); in eager it re-casts only when the key moved
                _, w16, wt16, b32 = cache
                w16.copy_(weight.detach())
                if wt16 is not None:
                    wt16.copy_(w16.t())
                if bias is not None:
                    if b32 is None:
                        b32 = bias.detach().float().contiguous()
                    else:
                        b32.copy_(bias.detach())
                weight._dfa_cast_cache = (key, w16, wt16, b32)
            else:
                _, w16, wt16, b32 = cache
                if bias is not None and b32 is None:
                    b32 = bias.detach().float().contiguous()
        if bias is None:
            b32 = None
        # shape dispatch (measured, profiles/optimization_r2.md): our gemm2
        # beats hipBLASLt on the square 768-ish shapes; the library wins the
        # fat no-bias ones (K or COL >= ~2k)
        if b32 is None and fat:
            out = torch.matmul(x2d, w16.t())
        else:
            out = ext.gemm2(x2d, w16, b32, None)
        ctx.save_for_backward(x2d, w16, wt16)
        ctx.has_bias = bias is not None
        ctx.x_shape = x.shape
        ctx.wp, ctx.bp = weight, bias
        return out.view(*x.shape[:-1], w16.shape[0])

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        x2d, w16, wt16 = ctx.saved_tensors
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        if w16.shape[0] > 1024 or w16.shape[1] > 1024:
            dx = torch.matmul(dy2d, w16)  # dY @ W, library wins fat shapes
        else:
            dx = ext.gemm2(dy2d, wt16, None, None)
        # direct accumulation (parallel/optim.py): under the flat optimizer
        # the kernels atomically add into the pre-zeroed flat .grad views and
        # autograd sees None — no zeros() fill, no AccumulateGrad add
        dw = db = None
        wp, bp = ctx.wp, ctx.bp
        if ctx.needs_input_grad[1]:
            if getattr(wp, "_dfa_w16", None) is not None and wp.grad is not None:
                ext.wgrad(dy2d, x2d, out=wp.grad)
            else:
                dw = ext.wgrad(dy2d, x2d)  # (COL, K) fp32
        if ctx.has_bias and ctx.needs_input_grad[2]:
            if getattr(bp, "_dfa_w16", None) is not None and bp.grad is not None:
                ext.colsum(dy2d, out=bp.grad)
            else:
                db = ext.colsum(dy2d)
        return dx.view(ctx.x_shape), dw, db


def linear_usable(x, weight) -> bool:
    n = x.numel() // x.shape[-1]
    K, COL = weight.shape[1], weight.shape[0]
    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and n % 128 == 0
        and K % 64 == 0
        and COL % 128 == 0
    )


def fused_linear(x, weight, bias=None):
    """F.linear with custom MFMA kernels where the geometry allows."""
    if linear_usable(x, weight):
        return _LinearBf16.apply(x, weight, bias)
    return torch.nn.functional.linear(x, weight, bias)


class _FlashAttention(torch.autograd.Function):
    """Fused attention, head_dim 64, (B, L, H*64) layout (csrc/flash_attn.hip):
    masked softmax + dropout + PV in one kernel, FA2-style two-pass backward.
    `bias` is the T5 additive position bias, fp32, TRANSPOSED key-major
    (H, Lk, Lq) — see models/t5.py _flash_bias_T."""

    @staticmethod
    def forward(ctx, q, k, v, H, valid, bias, scale, causal, dropout_p,
                bias_accum=None):
        ext = load_ext(required=True)
        seed = _next_seed() if dropout_p > 0 else 0
        # q/k/v may be row-strided slices of a fused QKV buffer — the
        # kernels take a row stride, so no contiguous() copies here
        O, lse = ext.flash_attn_fwd(q, k, v, H, valid, bias,
                                    scale, causal, dropout_p, seed)
        ctx.save_for_backward(q, k, v, O, lse)
        ctx.meta = (H, valid, bias, scale, causal, dropout_p, seed, bias_accum)
        ctx.need_dbias = bias is not None and ctx.needs_input_grad[5]
        return O

    @staticmethod
    def backward(ctx, dO):
        ext = load_ext(required=True)
        q, k, v, O, lse = ctx.saved_tensors
        H, valid, bias, scale, causal, dropout_p, seed, bias_accum = ctx.meta
        outs = ext.flash_attn_bwd(
            dO.contiguous(), q, k, v, O, lse,
            H, valid, bias, scale, causal, dropout_p, seed, ctx.need_dbias, False,
            bias_accum,
        )
        dbias = outs[3] if (ctx.need_dbias and bias_accum is None) else None
        return (outs[0], outs[1], outs[2], None, None, dbias, None, None, None,
                None)


def flash_attention(q, k, v, num_heads, valid=None, bias=None, scale=1.0,
                    causal=False, dropout_p=0.0, bias_accum=None):
    return _FlashAttention.apply(q, k, v, num_heads, valid, bias, scale, causal,
                                 dropout_p, bias_accum)


def flash_usable(x, L, Lk=None) -> bool:
    return (
        x.is_cuda
        and x.dtype == torch.bfloat16
        and L % 64 == 0
        and (Lk is None or Lk == L)
    )


class _RMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = load_ext(required=True)
        wf = weight.float().contiguous()
        x = x.contiguous()
        y, rstd = ext.rmsnorm_fwd(x, wf, eps)
        ctx.save_for_backward(x, wf, rstd)
        ctx.wp = weight
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        x, wf, rstd = ctx.saved_tensors
        dy = dy.contiguous()
        dx = ext.rmsnorm_bwd(dy, x, wf, rstd)
        wp = ctx.wp
        if (getattr(wp, "_dfa_w16", None) is not None and wp.grad is not None
                and ctx.needs_input_grad[1]):
            # direct accumulation into the flat .grad view (parallel/optim.py)
            ext.rmsnorm_wgrad(dy, x, rstd, out=wp.grad)
            return dx, None, None
        dgamma = ext.rmsnorm_wgrad(dy, x, rstd)
        return dx, dgamma, None


def rms_norm(x, weight, eps: float = 1e-6):
    """T5LayerNorm: no mean subtraction, no bias."""
    if x.is_cuda:
        return _RMSNorm.apply(x, weight, eps)
    v = x.float()
    y = v * torch.rsqrt(v.pow(2).mean(-1, keepdim=True) + eps)
    return (y * weight.float()).to(x.dtype)


def masked_softmax(S: torch.Tensor, valid: Optional[torch.Tensor], scale: float):
    # index 1 (the post-dropout output) is the differentiable one; with
    # dropout 0 it aliases the probabilities
    return masked_softmax_dropout(S, valid, scale, 0.0)[1]


class _EmbeddingLookup(torch.autograd.Function):
    """nn.Embedding fwd (torch gather) with a custom scatter-add backward
    (csrc/transformer_kernels.hip embed_scatter): replaces torch's
    radix-sort + segment-reduce + scatter stack in the grad path."""

    @staticmethod
    def forward(ctx, indices, weight, padding_idx):
        ctx.save_for_backward(indices)
        ctx.num_rows = weight.shape[0]
        ctx.w_dtype = weight.dtype
        ctx.padding_idx = padding_idx if padding_idx is not None else -1
        ctx.wp = weight
        return torch.nn.functional.embedding(indices, weight, padding_idx)

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        (indices,) = ctx.saved_tensors
        wp = ctx.wp
        if (getattr(wp, "_dfa_w16", None) is not None and wp.grad is not None
                and ctx.w_dtype == torch.float32):
            # scatter straight into the flat fp32 .grad view (pre-zeroed):
            # skips a (V, D) zeros + full-table AccumulateGrad add
            ext.embed_scatter(dy.contiguous(), indices.reshape(-1).contiguous(),
                              ctx.num_rows, ctx.padding_idx, out=wp.grad)
            return None, None, None
        dw = ext.embed_scatter(dy.contiguous(), indices.reshape(-1).contiguous(),
                               ctx.num_rows, ctx.padding_idx)
        return None, dw.to(ctx.w_dtype), None


def embedding_lookup(indices, weight, padding_idx=None):
    if indices.is_cuda:
        return _EmbeddingLookup.apply(indices, weight, padding_idx)
    return torch.nn.functional.embedding(indices, weight, padding_idx)


class _LayerNormResDropout(torch.autograd.Function):
    """y = LayerNorm(dropout(h) + residual) in one kernel per direction
    (csrc/transformer_kernels.hip ln_res_dropout_*): the transformer
    residual pattern with stateless dropout (mask regenerated from the
    seed in backward — only (h, res, seed) are kept)."""

    @staticmethod
    def forward(ctx, h, res, weight, bias, dropout_p, eps):
        ext = load_ext(required=True)
        wf = weight.float().contiguous()
        bf = bias.float().contiguous()
        h = h.contiguous()
        res = res.contiguous()
        seed = _next_seed() if dropout_p > 0 else 0
        y, mean, rstd = ext.ln_res_dropout_fwd(h, res, wf, bf, eps, dropout_p, seed)
        ctx.save_for_backward(h, res, wf, mean, rstd)
        ctx.meta = (dropout_p, seed)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        h, res, wf, mean, rstd = ctx.saved_tensors
        p, seed = ctx.meta
        dy = dy.contiguous()
        # bwd also reconstructs z = dropout(h)+res once, so the column
        # reduction runs the plain (hash-free) LN wgrad kernel
        dz, dh, z = ext.ln_res_dropout_bwd(dy, h, res, wf, mean, rstd, p, seed)
        dgamma, dbeta = ext.layernorm_wgrad(dy, z, mean, rstd)
        return dh, dz, dgamma, dbeta, None, None


def layer_norm_res_dropout(h, res, weight, bias, dropout_p=0.0, eps=1e-5):
    """Fused LayerNorm(dropout(h) + res); falls back to composition on CPU
    or when D is not a multiple of 256."""
    if h.is_cuda and h.shape[-1] % 256 == 0:
        return _LayerNormResDropout.apply(h, res, weight, bias, dropout_p, eps)
    z = torch.nn.functional.dropout(h, dropout_p) if dropout_p > 0 else h
    z = z + res
    return layer_norm(z, weight, bias, eps) if z.is_cuda else (
        torch.nn.functional.layer_norm(
            z.float(), (z.shape[-1],), weight.float(), bias.float(), eps
        ).to(z.dtype)
    )


class _QKVLinear(torch.autograd.Function):
    """Fused QKV projection: one (N, 3D) gemm2 / wgrad per layer instead of
    three 768-column calls (a 64x6-block grid underfills 256 CUs; the
    64x18 fused grid balances, and one K=8192 wgrad replaces three)."""

    @staticmethod
    def forward(ctx, x, wq, wk, wv, bq, bk, bv):
        ext = load_ext(required=True)
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        has_bias = bq is not None
        # bias-free q/k/v defined back-to-back (T5) sit ADJACENT in the flat
        # optimizer's bf16 shadow: the packed (3D, K) weight is a zero-copy
        # view of the shadow buffer — no per-step packing or cast kernels
        sq, sk, sv = _shadow_w16(wq), _shadow_w16(wk), _shadow_w16(wv)
        if (
            sq is not None and sk is not None and sv is not None
            and sk.data_ptr() == sq.data_ptr() + 2 * sq.numel()
            and sv.data_ptr() == sk.data_ptr() + 2 * sk.numel()
        ):
            base, off = wq._dfa_w16_base, wq._dfa_w16_off
            n3 = sq.numel() + sk.numel() + sv.numel()
            w16 = base[off : off + n3].view(3 * wq.shape[0], wq.shape[1])
            b32 = None
            if has_bias:
                cache = getattr(wq, "_dfa_qkv_cache", None)
                key = (bq._version, bk._version, bv._version, _weights_epoch[0])
                if cache is None or CAPTURE_REFRESH[0] or cache[0] != key:
                    b32 = torch.cat([bq.detach(), bk.detach(), bv.detach()]).float()
                    wq._dfa_qkv_cache = (key, b32)
                else:
                    b32 = cache[1]
        else:
            cache = getattr(wq, "_dfa_qkv_cache", None)
            key = (wq._version, wk._version, wv._version, _weights_epoch[0])
            if cache is None:
                wcat = torch.cat([wq.detach(), wk.detach(), wv.detach()])
                w16 = wcat.to(torch.bfloat16).contiguous()
                b32 = (torch.cat([bq.detach(), bk.detach(), bv.detach()]).float().contiguous()
                       if has_bias else None)
                wq._dfa_qkv_cache = (key, w16, b32)
            elif CAPTURE_REFRESH[0] or cache[0] != key:
                _, w16, b32 = cache
                D = wq.shape[0]
                w16[:D].copy_(wq.detach())
                w16[D:2 * D].copy_(wk.detach())
                w16[2 * D:].copy_(wv.detach())
                if has_bias:
                    b32[:D].copy_(bq.detach())
                    b32[D:2 * D].copy_(bk.detach())
                    b32[2 * D:].copy_(bv.detach())
                wq._dfa_qkv_cache = (key, w16, b32)
            else:
                _, w16, b32 = cache
        out = ext.gemm2(x2d, w16, b32, None)
        ctx.save_for_backward(x2d, w16)
        ctx.has_bias = has_bias
        ctx.x_shape = x.shape
        ctx.wparams = (wq, wk, wv)
        return out.view(*x.shape[:-1], w16.shape[0])

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        x2d, w16 = ctx.saved_tensors
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = torch.matmul(dy2d, w16)  # (N, 3D) @ (3D, D): fat K, library wins
        D = w16.shape[0] // 3
        wq, wk, wv = ctx.wparams
        # q/k/v grads adjacent in the flat grad buffer (T5, bias-free):
        # ONE wgrad accumulates the packed (3D, K) grad straight into the
        # flat .grad region — no zeros, no slicing, no AccumulateGrad adds
        if (
            not ctx.has_bias
            and getattr(wq, "_dfa_w16", None) is not None and wq.grad is not None
            and wk.grad is not None and wv.grad is not None
            and wk.grad.data_ptr() == wq.grad.data_ptr() + 4 * wq.numel()
            and wv.grad.data_ptr() == wk.grad.data_ptr() + 4 * wk.numel()
        ):
            n3 = wq.numel() + wk.numel() + wv.numel()
            g3 = wq._dfa_gbase[wq._dfa_goff : wq._dfa_goff + n3].view(3 * D, -1)
            ext.wgrad(dy2d, x2d, out=g3)
            return (dx.view(ctx.x_shape), None, None, None, None, None, None)
        dw = ext.wgrad(dy2d, x2d)  # (3D, K) fp32
        if ctx.has_bias:
            db = ext.colsum(dy2d)
            return (dx.view(ctx.x_shape), dw[:D], dw[D:2 * D], dw[2 * D:],
                    db[:D], db[D:2 * D], db[2 * D:])
        return (dx.view(ctx.x_shape), dw[:D], dw[D:2 * D], dw[2 * D:],
                None, None, None)


def fused_qkv(x, wq, wk, wv, bq=None, bk=None, bv=None):
    """Returns the fused (B, L, 3D) projection, or None if the geometry
    doesn't fit the custom GEMMs (caller falls back to per-projection)."""
    if linear_usable(x, wq) and wq.shape[0] % 128 == 0:
        return _QKVLinear.apply(x, wq, wk, wv, bq, bk, bv)
    return None


class _KVLinear(torch.autograd.Function):
    """Fused bias-free K/V projection for cross-attention (both read the
    SAME encoder states): one (N, 2D) gemm2/wgrad per layer instead of two
    768-column calls. With the flat optimizer the packed (2D, K) weight is a
    zero-copy view (k/v adjacent), and the packed grad accumulates directly
    into the flat .grad region (one wgrad, no AccumulateGrad adds)."""

    @staticmethod
    def forward(ctx, x, wk, wv):
        ext = load_ext(required=True)
        x2d = x.reshape(-1, x.shape[-1]).contiguous()
        sk, sv = _shadow_w16(wk), _shadow_w16(wv)
        if (
            sk is not None and sv is not None
            and sv.data_ptr() == sk.data_ptr() + 2 * sk.numel()
        ):
            base, off = wk._dfa_w16_base, wk._dfa_w16_off
            n2 = sk.numel() + sv.numel()
            w16 = base[off : off + n2].view(2 * wk.shape[0], wk.shape[1])
        else:
            cache = getattr(wk, "_dfa_kv_cache", None)
            key = (wk._version, wv._version, _weights_epoch[0])
            if cache is None:
                w16 = torch.cat([wk.detach(), wv.detach()]).to(torch.bfloat16).contiguous()
                wk._dfa_kv_cache = (key, w16)
            elif CAPTURE_REFRESH[0] or cache[0] != key:
                _, w16 = cache
                D = wk.shape[0]
                w16[:D].copy_(wk.detach())
                w16[D:].copy_(wv.detach())
                wk._dfa_kv_cache = (key, w16)
            else:
                _, w16 = cache
        out = ext.gemm2(x2d, w16, None, None)
        ctx.save_for_backward(x2d, w16)
        ctx.x_shape = x.shape
        ctx.wparams = (wk, wv)
        return out.view(*x.shape[:-1], w16.shape[0])

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        x2d, w16 = ctx.saved_tensors
        dy2d = dy.reshape(-1, dy.shape[-1]).contiguous()
        dx = torch.matmul(dy2d, w16)  # (N, 2D) @ (2D, D): fat K, library wins
        D = w16.shape[0] // 2
        wk, wv = ctx.wparams
        if (
            getattr(wk, "_dfa_w16", None) is not None and wk.grad is not None
            and wv.grad is not None
            and wv.grad.data_ptr() == wk.grad.data_ptr() + 4 * wk.numel()
        ):
            n2 = wk.numel() + wv.numel()
            g2 = wk._dfa_gbase[wk._dfa_goff : wk._dfa_goff + n2].view(2 * D, -1)
            ext.wgrad(dy2d, x2d, out=g2)
            return dx.view(ctx.x_shape), None, None
        dw = ext.wgrad(dy2d, x2d)
        return dx.view(ctx.x_shape), dw[:D], dw[D:]


def fused_kv(x, wk, wv):
    """Fused (B, L, 2D) cross-attention K/V projection, or None when the
    geometry doesn't fit the custom GEMMs."""
    if linear_usable(x, wk) and wk.shape[0] % 128 == 0:
        return _KVLinear.apply(x, wk, wv)
    return None


class _FlashAttentionKV(torch.autograd.Function):
    """Flash attention with a dense Q and the FUSED (B, L, 2D) K/V
    projection (cross-attention): k/v are strided slices forward, backward
    writes dk/dv into one (B, L, 2D) buffer (flash_attn_bwd kv_fused)."""

    @staticmethod
    def forward(ctx, q, kv, H, valid, scale, dropout_p):
        ext = load_ext(required=True)
        D = kv.shape[-1] // 2
        k, v = kv[..., :D], kv[..., D:]
        seed = _next_seed() if dropout_p > 0 else 0
        O, lse = ext.flash_attn_fwd(q, k, v, H, valid, None, scale, False,
                                    dropout_p, seed)
        ctx.save_for_backward(q, kv, O, lse)
        ctx.meta = (H, valid, scale, dropout_p, seed)
        return O

    @staticmethod
    def backward(ctx, dO):
        ext = load_ext(required=True)
        q, kv, O, lse = ctx.saved_tensors
        H, valid, scale, dropout_p, seed = ctx.meta
        D = kv.shape[-1] // 2
        k, v = kv[..., :D], kv[..., D:]
        dq, dkv = ext.flash_attn_bwd(
            dO.contiguous(), q, k, v, O, lse, H, valid, None, scale, False,
            dropout_p, seed, False, False, None, kv_fused=True,
        )
        return dq, dkv, None, None, None, None


def flash_attention_kv(q, kv, num_heads, valid=None, scale=1.0, dropout_p=0.0):
    return _FlashAttentionKV.apply(q, kv, num_heads, valid, scale, dropout_p)


class _DropoutAdd(torch.autograd.Function):
    """out = res + dropout(h) in one kernel (T5 pre-norm residuals);
    d(res) = dy passes through with no kernel."""

    @staticmethod
    def forward(ctx, h, res, p):
        ext = load_ext(required=True)
        seed = _next_seed() if p > 0 else 0
        ctx.meta = (p, seed)
        return ext.dropout_add_fwd(h.contiguous(), res.contiguous(), p, seed)

    @staticmethod
    def backward(ctx, dy):
        p, seed = ctx.meta
        if p == 0:
            return dy, dy, None
        ext = load_ext(required=True)
        dy = dy.contiguous()
        return ext.dropout_add_bwd(dy, p, seed), dy, None


def dropout_add(h, res, p=0.0, training=True):
    """res + dropout(h); fused on GPU, torch composition elsewhere."""
    if not training:
        p = 0.0
    if h.is_cuda and h.numel() % (4 if h.dtype == torch.float32 else 8) == 0:
        return _DropoutAdd.apply(h, res, p)
    z = torch.nn.functional.dropout(h, p) if p > 0 else h
    return res + z


class _FlashAttentionQKV(torch.autograd.Function):
    """Flash attention taking the FUSED (B, L, 3D) QKV projection as one
    input: forward slices q/k/v as strided views (the kernels take row
    strides), backward writes dq/dk/dv into ONE (B, L, 3D) buffer — no
    slice-backward zero+scatter work in autograd."""

    @staticmethod
    def forward(ctx, qkv, H, valid, bias, scale, causal, dropout_p,
                bias_accum=None):
        ext = load_ext(required=True)
        D = qkv.shape[-1] // 3
        q, k, v = qkv[..., :D], qkv[..., D:2 * D], qkv[..., 2 * D:]
        seed = _next_seed() if dropout_p > 0 else 0
        O, lse = ext.flash_attn_fwd(q, k, v, H, valid, bias, scale, causal,
                                    dropout_p, seed)
        ctx.save_for_backward(qkv, O, lse)
        ctx.meta = (H, valid, bias, scale, causal, dropout_p, seed, bias_accum)
        ctx.need_dbias = bias is not None and ctx.needs_input_grad[3]
        return O

    @staticmethod
    def backward(ctx, dO):
        ext = load_ext(required=True)
        qkv, O, lse = ctx.saved_tensors
        H, valid, bias, scale, causal, dropout_p, seed, bias_accum = ctx.meta
        D = qkv.shape[-1] // 3
        q, k, v = qkv[..., :D], qkv[..., D:2 * D], qkv[..., 2 * D:]
        outs = ext.flash_attn_bwd(
            dO.contiguous(), q, k, v, O, lse,
            H, valid, bias, scale, causal, dropout_p, seed, ctx.need_dbias, True,
            bias_accum,
        )
        # bias_accum set: the dq kernel atomics landed in the SHARED buffer
        # (T5's 24 layers share one position bias) — no per-layer dBias
        dbias = outs[1] if (ctx.need_dbias and bias_accum is None) else None
        return outs[0], None, None, dbias, None, None, None, None


def flash_attention_qkv(qkv, num_heads, valid=None, bias=None, scale=1.0,
                        causal=False, dropout_p=0.0, bias_accum=None):
    return _FlashAttentionQKV.apply(qkv, num_heads, valid, bias, scale, causal,
                                    dropout_p, bias_accum)


class _LMHeadCE(torch.autograd.Function):
    """K21: fused LM-head GEMM + cross-entropy over the 32100-token vocab
    (csrc/lmhead_ce.hip). Forward streams vocab tiles with online
    logsumexp — logits are never materialized (the reference call site,
    CodeT5/models.py:140-149, materializes b*512*32100 fp32 twice).
    Backward recomputes tiles into scaled bf16 dlogits, then dh/dW are two
    library GEMMs. Semantics == F.cross_entropy(logits.float(), targets,
    ignore_index=-100) with logits = scale * h @ W^T (mean over valid)."""

    @staticmethod
    def forward(ctx, h, weight, targets, scale):
        ext = load_ext(required=True)
        K = h.shape[-1]
        h2d = h.reshape(-1, K).to(torch.bfloat16).contiguous()
        V = weight.shape[0]
        Vp = (V + 127) // 128 * 128
        # padded bf16 vocab-weight cache (same invalidation protocol as
        # _LinearBf16: parameter version + weights epoch + capture refresh)
        cache = getattr(weight, "_dfa_vocab_cache", None)
        key = (weight._version, _weights_epoch[0])
        if cache is None:
            Wp = torch.zeros(Vp, K, dtype=torch.bfloat16, device=weight.device)
            src = _shadow_w16(weight)
            Wp[:V].copy_(src if src is not None else weight.detach())
            weight._dfa_vocab_cache = (key, Wp)
        elif CAPTURE_REFRESH[0] or cache[0] != key:
            _, Wp = cache
            # refresh from the optimizer's bf16 shadow when present: a
            # bf16->bf16 copy is 2/3 the traffic of the fp32 cast at V=32100
            src = _shadow_w16(weight)
            Wp[:V].copy_(src if src is not None else weight.detach())
            weight._dfa_vocab_cache = (key, Wp)
        else:
            _, Wp = cache
        tgt = targets.reshape(-1).to(torch.int32)
        tgt = torch.where(tgt == -100, tgt.new_full((), -1), tgt).contiguous()
        loss_rows, lse = ext.lmhead_ce_fwd(h2d, Wp, tgt, float(scale), V)
        n_valid = (tgt >= 0).sum().to(torch.float32).clamp(min=1.0)
        loss = loss_rows.sum() / n_valid
        ctx.save_for_backward(h2d, Wp, tgt, lse, n_valid)
        ctx.scale = float(scale)
        ctx.V = V
        ctx.h_shape = h.shape
        ctx.h_dtype = h.dtype
        ctx.w_dtype = weight.dtype
        return loss

    @staticmethod
    def backward(ctx, grad):
        ext = load_ext(required=True)
        h2d, Wp, tgt, lse, n_valid = ctx.saved_tensors
        # fold the chain scale into gscale once: it multiplies both dh and dW
        g = grad.float() * ctx.scale / n_valid
        gscale = torch.where(tgt >= 0, g, torch.zeros_like(g)).contiguous()
        dlogits = ext.lmhead_ce_bwd(h2d, Wp, tgt, lse, gscale, ctx.scale, ctx.V)
        dh = torch.matmul(dlogits, Wp)            # (M, K) bf16
        dw = torch.matmul(dlogits.t(), h2d)       # (Vp, K) bf16
        return (
            dh.view(ctx.h_shape).to(ctx.h_dtype),
            dw[: ctx.V].to(ctx.w_dtype),
            None,
            None,
        )


def lmhead_ce_usable(h, weight) -> bool:
    if not (h.is_cuda and load_ext() is not None):
        return False
    M = h.numel() // h.shape[-1]
    return M % 128 == 0 and h.shape[-1] % 64 == 0


def lmhead_cross_entropy(h, weight, targets, scale: float = 1.0):
    """Mean CE over rows where target != -100, logits = scale * h @ W^T."""
    if lmhead_ce_usable(h, weight):
        return _LMHeadCE.apply(h, weight, targets, scale)
    logits = torch.nn.functional.linear(h.reshape(-1, h.shape[-1]) * scale, weight)
    return torch.nn.functional.cross_entropy(
        logits.float(), targets.reshape(-1), ignore_index=-100
    )


class _ReluDropout(torch.autograd.Function):
    """Fused ReLU + dropout (T5 FFN inner activation): stateless mask from
    the seed, only the pre-ReLU input saved."""

    @staticmethod
    def forward(ctx, x, p, training):
        ext = load_ext(required=True)
        seed = _next_seed() if (training and p > 0) else 0
        pp = p if training else 0.0
        xc = x.contiguous()
        out = ext.relu_dropout_fwd(xc, pp, seed)
        ctx.save_for_backward(xc)
        ctx.p = pp
        ctx.seed = seed
        return out

    @staticmethod
    def backward(ctx, dy):
        ext = load_ext(required=True)
        (x,) = ctx.saved_tensors
        dx = ext.relu_dropout_bwd(dy.contiguous(), x, ctx.p, ctx.seed)
        return dx, None, None


def relu_dropout(x, p=0.0, training=True):
    if x.is_cuda:
        return _ReluDropout.apply(x, p, training)
    h = torch.relu(x)
    return torch.nn.functional.dropout(h, p) if (training and p > 0) else h
