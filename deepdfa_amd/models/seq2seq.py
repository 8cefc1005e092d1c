"""CodeBERT-style encoder-decoder for generation tasks.

Capability parity: reference CodeT5/models.py:195-295 (Seq2Seq: RoBERTa
encoder + transformer decoder, tanh(dense) projection, lm_head tied to the
encoder word embeddings, shifted CE over active target positions, beam
search decode) and the Beam object (:298-408). Built MI355X-first on our
own modules instead of torch.nn.TransformerDecoder: the decoder layers use
the same fused attention-softmax / bias-GELU / MFMA-linear ops as the
RoBERTa encoder, with causal self-attention plus cross-attention over the
encoder memory, and beam search is batched over the whole beam (one kernel
launch set per step) instead of a per-example python Beam loop.
"""

from __future__ import annotations

import math

import torch
import torch.nn as nn

from ..ops.transformer import fused_linear, masked_softmax_dropout
from .roberta import LayerNorm, RobertaConfig, RobertaModel, init_roberta_weights


class _MHA(nn.Module):
    """Multi-head attention over (B, L, D); `causal` for decoder self-attn,
    `kv` for cross-attention over the encoder memory."""

    def __init__(self, cfg: RobertaConfig):
        super().__init__()
        D = cfg.hidden_size
        self.H = cfg.num_attention_heads
        self.d = D // self.H
        self.query = nn.Linear(D, D)
        self.key = nn.Linear(D, D)
        self.value = nn.Linear(D, D)
        self.dropout_p = cfg.attention_probs_dropout_prob

    def forward(self, x, kv=None, kv_valid=None, causal=False):
        kv = x if kv is None else kv
        B, Lq, D = x.shape
        Lk = kv.shape[1]
        q = self.query(x).view(B, Lq, self.H, self.d).transpose(1, 2)
        k = self.key(kv).view(B, Lk, self.H, self.d).transpose(1, 2)
        v = self.value(kv).view(B, Lk, self.H, self.d).transpose(1, 2)
        scores = torch.matmul(q, k.transpose(-1, -2))
        p = self.dropout_p if self.training else 0.0
        _, probs = masked_softmax_dropout(
            scores, kv_valid, 1.0 / math.sqrt(self.d), p, causal=causal
        )
        out = torch.matmul(probs, v)
        return out.transpose(1, 2).reshape(B, Lq, D)


class Seq2SeqDecoderLayer(nn.Module):
    def __init__(self, cfg: RobertaConfig):
        super().__init__()
        D = cfg.hidden_size
        self.self_attn = _MHA(cfg)
        self.cross_attn = _MHA(cfg)
        self.self_out = nn.Linear(D, D)
        self.cross_out = nn.Linear(D, D)
        self.norm1 = LayerNorm(D, cfg.layer_norm_eps)
        self.norm2 = LayerNorm(D, cfg.layer_norm_eps)
        self.norm3 = LayerNorm(D, cfg.layer_norm_eps)
        self.ff1 = nn.Linear(D, cfg.intermediate_size)
        self.ff2 = nn.Linear(cfg.intermediate_size, D)
        self.dropout = nn.Dropout(cfg.hidden_dropout_prob)

    def forward(self, x, memory, memory_valid):
        h = self.self_attn(x, causal=True)
        x = self.norm1(x + self.dropout(self.self_out(h)))
        h = self.cross_attn(x, kv=memory, kv_valid=memory_valid)
        x = self.norm2(x + self.dropout(self.cross_out(h)))
        h = self.ff2(torch.nn.functional.gelu(fused_linear(x, self.ff1.weight, self.ff1.bias)))
        return self.norm3(x + self.dropout(h))


class Seq2Seq(nn.Module):
    """encoder: RobertaModel (ours). decoder: stack of Seq2SeqDecoderLayer.
    lm_head weight tied to encoder.embeddings.word_embeddings."""

    def __init__(self, config: RobertaConfig, num_decoder_layers: int = 6,
                 beam_size: int = 10, max_length: int = 64,
                 sos_id: int = 0, eos_id: int = 2, encoder: RobertaModel = None):
        super().__init__()
        self.config = config
        self.encoder = encoder if encoder is not None else RobertaModel(config)
        self.decoder = nn.ModuleList(
            Seq2SeqDecoderLayer(config) for _ in range(num_decoder_layers)
        )
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.lm_head.weight = self.encoder.embeddings.word_embeddings.weight
        self.beam_size = beam_size
        self.max_length = max_length
        self.sos_id = sos_id
        self.eos_id = eos_id
        init_roberta_weights(self.decoder)
        init_roberta_weights(self.dense)

    def _decode(self, target_ids, memory, memory_valid):
        x = self.encoder.embeddings(target_ids)
        if x.is_cuda and torch.is_autocast_enabled():
            x = x.to(torch.bfloat16)
        for lyr in self.decoder:
            x = lyr(x, memory, memory_valid)
        return self.lm_head(torch.tanh(self.dense(x)))

    def forward(self, source_ids, source_mask=None, target_ids=None, target_mask=None):
        """Training: returns (loss, loss*num_active, num_active) like the
        reference (:258-263). Inference (target_ids None): beam-search
        prediction ids (B, beam, max_length)."""
        if source_mask is None:
            source_mask = source_ids.ne(self.config.pad_token_id)
        memory, _ = self.encoder(source_ids, attention_mask=source_mask)
        memory_valid = source_mask.sum(dim=1).to(torch.int32)
        if target_ids is None:
            return self._beam(memory, memory_valid)
        lm_logits = self._decode(target_ids, memory, memory_valid)
        if target_mask is None:
            target_mask = target_ids.ne(self.config.pad_token_id)
        active = target_mask[..., 1:].ne(0).reshape(-1)
        shift_logits = lm_logits[..., :-1, :].reshape(-1, lm_logits.shape[-1])[active]
        shift_labels = target_ids[..., 1:].reshape(-1)[active]
        loss = nn.functional.cross_entropy(shift_logits.float(), shift_labels)
        return loss, loss * active.sum(), active.sum()

    @torch.no_grad()
    def _beam(self, memory, memory_valid):
        """Batched beam search: the (B*beam) hypotheses decode together —
        one fused kernel set per step across the whole beam frontier."""
        B, K = memory.shape[0], self.beam_size
        dev = memory.device
        mem = memory.repeat_interleave(K, dim=0)
        mv = memory_valid.repeat_interleave(K, dim=0)
        seqs = torch.full((B * K, 1), self.sos_id, dtype=torch.long, device=dev)
        scores = torch.full((B, K), float("-inf"), device=dev)
        scores[:, 0] = 0.0
        done = torch.zeros(B * K, dtype=torch.bool, device=dev)
        for _ in range(self.max_length - 1):
            logits = self._decode(seqs, mem, mv)[:, -1, :].float()
            lp = torch.log_softmax(logits, dim=-1)
            # frozen finished hypotheses: only eos continues, at zero cost
            lp[done] = float("-inf")
            lp[done, self.eos_id] = 0.0
            V = lp.shape[-1]
            total = scores.view(B * K, 1) + lp
            top_s, top_i = total.view(B, K * V).topk(K, dim=-1)
            beam_src = top_i // V
            tok = (top_i % V).view(B * K, 1)
            gather = (beam_src + torch.arange(B, device=dev).view(B, 1) * K).view(-1)
            seqs = torch.cat([seqs[gather], tok], dim=1)
            done = done[gather] | (tok.view(-1) == self.eos_id)
            scores = top_s
            if bool(done.all()):
                break
        out = seqs[:, 1:]
        pad = self.max_length - out.shape[1]
        if pad > 0:
            out = torch.nn.functional.pad(out, (0, pad), value=0)
        # zero everything after (and including trailing pads past) eos
        eos = out == self.eos_id
        after = eos.cumsum(dim=1) - eos.to(torch.long) > 0
        out = out.masked_fill(after | eos, 0)
        return out.view(B, K, -1)
