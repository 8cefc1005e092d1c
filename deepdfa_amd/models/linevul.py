"""LineVul model: CodeBERT classifier, optionally combined with the
DeepDFA flow-GNN encoder.

Parity target: reference LineVul/linevul/linevul_model.py:1-69 —
`RobertaClassificationHead` (concat [CLS] 768 with the 256-d flow-GNN
embedding -> Linear(1024,768) -> tanh -> dropout -> Linear(768,2)) and
`Model.forward` (CE loss + softmax probs; attention-weights branch for
line-level scoring, :42-56)."""

from __future__ import annotations

from typing import Optional

import torch
from torch import nn

from .roberta import RobertaConfig, RobertaModel, init_roberta_weights


class RobertaClassificationHead(nn.Module):
    """linevul_model.py:6-24 semantics."""

    def __init__(self, config: RobertaConfig, extra_dim: int = 0):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size + extra_dim, config.hidden_size)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.out_proj = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, features, flowgnn_embed: Optional[torch.Tensor] = None):
        x = features[:, 0, :]  # [CLS]
        if flowgnn_embed is not None:
            x = torch.cat((x, flowgnn_embed.to(x.dtype)), dim=1)
        x = self.dropout(x)
        x = torch.tanh(self.dense(x))
        x = self.dropout(x)
        return self.out_proj(x)


class Model(nn.Module):
    """Combined LineVul(+DeepDFA) classifier (linevul_model.py:26-69)."""

    def __init__(
        self,
        encoder: Optional[RobertaModel] = None,
        flowgnn_encoder: Optional[nn.Module] = None,
        config: Optional[RobertaConfig] = None,
        tokenizer=None,
        args=None,
    ):
        super().__init__()
        if config is None:
            config = RobertaConfig()
        if encoder is None:
            encoder = RobertaModel(config)
            init_roberta_weights(encoder)
        self.encoder = encoder
        self.flowgnn_encoder = flowgnn_encoder
        extra = flowgnn_encoder.out_dim if flowgnn_encoder is not None else 0
        self.classifier = RobertaClassificationHead(config, extra_dim=extra)
        init_roberta_weights(self.classifier)
        self.tokenizer = tokenizer
        self.args = args

    def forward(
        self,
        input_ids: torch.Tensor,
        labels: Optional[torch.Tensor] = None,
        graphs=None,
        output_attentions: bool = False,
    ):
        attention_mask = input_ids.ne(1)
        if self.flowgnn_encoder is not None and graphs is not None:
            flowgnn_embed = self.flowgnn_encoder(graphs, {})
        else:
            flowgnn_embed = None
        hidden, attentions = self.encoder(
            input_ids, attention_mask=attention_mask, output_attentions=output_attentions
        )
        logits = self.classifier(hidden, flowgnn_embed)
        prob = torch.softmax(logits.float(), dim=-1)
        if labels is not None:
            loss = torch.nn.functional.cross_entropy(logits.float(), labels)
            if output_attentions:
                return loss, prob, attentions
            return loss, prob
        if output_attentions:
            return prob, attentions
        return prob
