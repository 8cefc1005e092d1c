"""CodeT5 DefectModel: T5 seq2seq backbone + defect classifier, optionally
combined with the DeepDFA flow-GNN.

Parity target: reference CodeT5/models.py:125-191 — get_t5_vec runs the
FULL encoder-decoder with labels=source_ids (teacher forcing every step),
pools the decoder's last hidden state at the final EOS position per row
(:138-149, eos-count check :145-146), concats the flow-GNN embedding
(:179-181) and classifies with Linear(768[+256], 2) + CE (:131-134,183-191).
"""

from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from .t5 import T5Config, T5ForConditionalGeneration


class DefectModel(nn.Module):
    def __init__(
        self,
        encoder: Optional[T5ForConditionalGeneration] = None,
        config: Optional[T5Config] = None,
        tokenizer=None,
        args=None,
        flowgnn_encoder: Optional[nn.Module] = None,
    ):
        super().__init__()
        if config is None:
            config = T5Config()
        self.config = config
        self.encoder = encoder if encoder is not None else T5ForConditionalGeneration(config)
        self.tokenizer = tokenizer
        self.args = args
        self.flowgnn_encoder = flowgnn_encoder
        extra = flowgnn_encoder.out_dim if flowgnn_encoder is not None else 0
        self.classifier = nn.Linear(config.d_model + extra, 2)

    def get_t5_vec(self, source_ids: torch.Tensor) -> torch.Tensor:
        attention_mask = source_ids.ne(self.config.pad_token_id)
        hidden = self.encoder(
            source_ids, attention_mask=attention_mask, labels=source_ids,
            output_hidden_only=True,
        )
        eos_mask = source_ids.eq(self.config.eos_token_id)
        if not source_ids.is_cuda and len(torch.unique(eos_mask.sum(1))) > 1:
            raise ValueError("All examples must have the same number of <eos> tokens.")
        # last-EOS gather with fixed shapes (the reference's boolean-index
        # form forces a host sync + data-dependent shape every forward;
        # equal-eos-count validation stays on the CPU path)
        b, L, d = hidden.shape
        idx = torch.where(
            eos_mask, torch.arange(L, device=source_ids.device), -1
        ).max(dim=1).values.clamp_(min=0)
        return hidden.gather(1, idx.view(b, 1, 1).expand(b, 1, d)).squeeze(1)

    def forward(self, source_ids, labels=None, graphs=None):
        vec = self.get_t5_vec(source_ids)
        if self.flowgnn_encoder is not None and graphs is not None:
            flowgnn_embed = self.flowgnn_encoder(graphs, {})
            vec = torch.cat((vec, flowgnn_embed.to(vec.dtype)), dim=1)
        logits = self.classifier(vec.float())
        prob = torch.softmax(logits, dim=-1)
        if labels is not None:
            loss = torch.nn.functional.cross_entropy(logits, labels)
            return loss, prob
        return prob


class RobertaClassificationHead(nn.Module):
    """Pair-classification head (reference CodeT5/models.py clone head):
    concat the two EOS vectors -> dense -> tanh -> out_proj."""

    def __init__(self, config: T5Config):
        super().__init__()
        self.dense = nn.Linear(config.d_model * 2, config.d_model)
        self.out_proj = nn.Linear(config.d_model, 2)

    def forward(self, x):
        x = x.reshape(-1, x.size(-1) * 2)
        x = torch.tanh(self.dense(x))
        return self.out_proj(x)


class CloneModel(nn.Module):
    """Clone detection (reference CodeT5/models.py:64-122): each example is
    a PAIR of functions; both run the full seq2seq EOS pooling; the pair of
    vectors feeds the classification head."""

    def __init__(self, encoder=None, config: Optional[T5Config] = None, tokenizer=None,
                 args=None, max_source_length: int = 512):
        super().__init__()
        if config is None:
            config = T5Config()
        self.config = config
        self.encoder = encoder if encoder is not None else T5ForConditionalGeneration(config)
        self.classifier = RobertaClassificationHead(config)
        self.tokenizer = tokenizer
        self.args = args
        self.max_source_length = (
            getattr(args, "max_source_length", max_source_length) if args else max_source_length
        )

    def get_t5_vec(self, source_ids):
        attention_mask = source_ids.ne(self.config.pad_token_id)
        hidden = self.encoder(source_ids, attention_mask=attention_mask, labels=source_ids,
                              output_hidden_only=True)
        eos_mask = source_ids.eq(self.config.eos_token_id)
        if not source_ids.is_cuda and len(torch.unique(eos_mask.sum(1))) > 1:
            raise ValueError("All examples must have the same number of <eos> tokens.")
        b, L, d = hidden.shape
        idx = torch.where(
            eos_mask, torch.arange(L, device=source_ids.device), -1
        ).max(dim=1).values.clamp_(min=0)
        return hidden.gather(1, idx.view(b, 1, 1).expand(b, 1, d)).squeeze(1)

    def forward(self, source_ids, labels=None):
        source_ids = source_ids.view(-1, self.max_source_length)
        vec = self.get_t5_vec(source_ids)
        logits = self.classifier(vec.float())
        prob = torch.softmax(logits, dim=-1)
        if labels is not None:
            loss = torch.nn.functional.cross_entropy(logits, labels)
            return loss, prob
        return prob
