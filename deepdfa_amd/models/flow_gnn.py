"""FlowGNN: gated-graph network over batched CFGs (the DeepDFA model).

Parity target: reference DDFA/code_gnn/models/flow_gnn/ggnn.py:22-109
(FlowGNNGGNNModule) and DGL's GatedGraphConv/GlobalAttentionPooling
semantics, rebuilt on the MI355X-native op set:

  4 x nn.Embedding(input_dim, 32) -> concat (N,128)     [K1 fused gather]
  5 x { m = spmm_sum(W h + b)                            [MFMA GEMM + K2]
        h = GRUCell(m, h) }                              [GEMMs + K3 gates]
  out = cat([h, feat_embed])  (N,256)                    [K4]
  pool: gate = Linear(256,1); segment-softmax; sum       [K5]
  head: [Linear(256,256)+ReLU]x2 + Linear(256,1)         [K6 MFMA]

With concat_all_absdf=True and hidden_dim=32 (config_ggnn.yaml) the
embedding dim and GGNN width are both 128, matching the reference's
`embedding_dim *= 4; hidden_dim *= 4` (ggnn.py:48-52).
"""

from __future__ import annotations

from typing import Dict, Optional

import torch
from torch import nn

from ..ops import attn_pool, embed4, gru_cell, spmm_sum
from ..ops.flowgnn import ggnn_fused
from .base_module import BaseModule

ALL_FEATS = ["api", "datatype", "literal", "operator"]


class GatedGraphConv(nn.Module):
    """DGL GatedGraphConv equivalent (n_etypes=1): per step,
    m_v = sum_{u->v}(W h_u + b) then h = GRUCell(m, h).
    The per-edge linear commutes with the sum, so we compute W h + b per
    node once and aggregate with the CSR segment-sum kernel (the bias is
    aggregated too, weighted by in-degree — exactly DGL's apply_edges
    semantics)."""

    def __init__(self, in_feats: int, out_feats: int, n_steps: int, n_etypes: int = 1):
        super().__init__()
        assert n_etypes == 1, "CFG-only build uses a single edge type"
        if in_feats > out_feats:
            raise ValueError("in_feats must be <= out_feats (DGL contract)")
        self.in_feats = in_feats
        self.out_feats = out_feats
        self.n_steps = n_steps
        self.linear = nn.Linear(out_feats, out_feats)
        self.gru = nn.GRUCell(out_feats, out_feats)
        self.reset_parameters()

    def reset_parameters(self):
        gain = nn.init.calculate_gain("relu")
        nn.init.xavier_normal_(self.linear.weight, gain=gain)
        nn.init.zeros_(self.linear.bias)
        self.gru.reset_parameters()

    def forward(self, graph, feat: torch.Tensor) -> torch.Tensor:
        h = feat
        if self.in_feats < self.out_feats:
            pad = torch.zeros(
                h.shape[0], self.out_feats - self.in_feats, dtype=h.dtype, device=h.device
            )
            h = torch.cat([h, pad], dim=1)
        # MI355X fast path: the whole unrolled loop as one C++-driven bf16
        # autograd node (MFMA GEMMs + fused gates). Engaged under bf16 /
        # autocast on GPU; the op-by-op path below remains the fp32 route
        # (and the CPU route) used by the numerics tests.
        if h.is_cuda and (
            h.dtype == torch.bfloat16 or torch.is_autocast_enabled()
        ) and self.out_feats % 64 == 0 and (4 * self.out_feats) % 128 == 0:
            return ggnn_fused(h, graph, self.linear, self.gru, self.n_steps)
        for _ in range(self.n_steps):
            wh = self.linear(h)
            m = spmm_sum(wh, graph)
            h = gru_cell(
                m,
                h,
                self.gru.weight_ih,
                self.gru.weight_hh,
                self.gru.bias_ih,
                self.gru.bias_hh,
            )
        return h


class GlobalAttentionPooling(nn.Module):
    """DGL GlobalAttentionPooling equivalent over batched-CFG segments."""

    def __init__(self, gate_nn: nn.Linear):
        super().__init__()
        self.gate_nn = gate_nn

    def forward(self, graph, feat: torch.Tensor) -> torch.Tensor:
        gate = self.gate_nn(feat).squeeze(-1)
        return attn_pool(feat, gate, graph)


class FlowGNNGGNNModule(BaseModule):
    """Constructor-compatible with the reference FlowGNNGGNNModule
    (ggnn.py:23-80): same arguments, same submodule names so state_dicts
    map 1:1 (all_embeddings.{api,...}.weight, ggnn.linear.*, ggnn.gru.*,
    pooling.gate_nn.*, output_layer.{0,2,4}.*)."""

    def __init__(
        self,
        feat: str = "_ABS_DATAFLOW_datatype_all_limitall_1000_limitsubkeys_1000",
        input_dim: int = 1002,
        hidden_dim: int = 32,
        n_steps: int = 5,
        num_output_layers: int = 3,
        label_style: str = "graph",
        concat_all_absdf: bool = True,
        encoder_mode: bool = False,
        **kwargs,
    ):
        super().__init__(label_style=label_style, **kwargs)
        if "_ABS_DATAFLOW" in feat:
            feat = "_ABS_DATAFLOW"
        self.feature_keys = {"feature": feat}
        self.input_dim = input_dim
        self.concat_all_absdf = concat_all_absdf
        self.encoder_mode = encoder_mode
        self.hparams.update(
            feat=feat,
            input_dim=input_dim,
            hidden_dim=hidden_dim,
            n_steps=n_steps,
            num_output_layers=num_output_layers,
            label_style=label_style,
            concat_all_absdf=concat_all_absdf,
            encoder_mode=encoder_mode,
        )

        embedding_dim = hidden_dim
        if concat_all_absdf:
            self.all_embeddings = nn.ModuleDict(
                {of: nn.Embedding(input_dim, embedding_dim) for of in ALL_FEATS}
            )
            embedding_dim *= len(ALL_FEATS)
            hidden_dim *= len(ALL_FEATS)
        else:
            self.embedding = nn.Embedding(input_dim, embedding_dim)

        self.ggnn = GatedGraphConv(
            in_feats=embedding_dim, out_feats=hidden_dim, n_steps=n_steps, n_etypes=1
        )
        output_in_size = embedding_dim + hidden_dim
        self.out_dim = output_in_size

        if label_style == "graph":
            self.pooling = GlobalAttentionPooling(nn.Linear(output_in_size, 1))

        if not encoder_mode:
            layers = []
            for i in range(num_output_layers):
                out_size = 1 if i == num_output_layers - 1 else output_in_size
                layers.append(nn.Linear(output_in_size, out_size))
                if i != num_output_layers - 1:
                    layers.append(nn.ReLU())
            self.output_layer = nn.Sequential(*layers)

    # -- forward -------------------------------------------------------------

    def _embed(self, graph) -> torch.Tensor:
        if self.concat_all_absdf:
            idx = torch.stack(
                [graph.ndata[f"_ABS_DATAFLOW_{of}"] for of in ALL_FEATS], dim=1
            )
            if torch.is_grad_enabled() and idx.is_cuda:
                # flat-optimizer fast path: zero-copy bf16 stack view of the
                # adjacent tables + direct scatter into the flat .grad region
                from ..ops.flowgnn import embed4_direct

                ws = [self.all_embeddings[of].weight for of in ALL_FEATS]
                out = embed4_direct(ws, idx)
                if out is not None:
                    return out
            tables = self._stacked_tables()
            return embed4(tables, idx)
        feat = graph.ndata[self.feature_keys["feature"]]
        return self.embedding(feat)

    def _stacked_tables(self) -> torch.Tensor:
        """(4, V, 32) stacked embedding tables, cached per weight version
        (the per-forward torch.stack was one more graph node per step);
        non-leaf during training so grads still flow to each table."""
        ws = [self.all_embeddings[of].weight for of in ALL_FEATS]
        if any(w.requires_grad for w in ws) and torch.is_grad_enabled():
            return torch.stack(ws)  # autograd path: must stay in the graph
        from ..ops.transformer import CAPTURE_REFRESH, _weights_epoch

        key = tuple(w._version for w in ws) + (_weights_epoch[0], str(ws[0].device))
        cache = getattr(self, "_dfa_tables_cache", None)
        if cache is not None and cache[1].device != ws[0].device:
            cache = None  # model moved after caching
        if cache is None:
            buf = torch.stack([w.detach() for w in ws])
            self._dfa_tables_cache = (key, buf)
        elif CAPTURE_REFRESH[0] or cache[0] != key:
            buf = cache[1]
            for i, w in enumerate(ws):
                buf[i].copy_(w.detach())
            self._dfa_tables_cache = (key, buf)
        else:
            buf = cache[1]
        return buf

    def forward(self, graph, extrafeats: Optional[Dict] = None) -> torch.Tensor:
        feat_embed = self._embed(graph)
        ggnn_out = self.ggnn(graph, feat_embed)
        # fused head (GPU bf16): concat + gate GEMV + segment-softmax pool in
        # one kernel each way; the 3-layer MLP as one more — the eager tail
        # was ~25 launch-floor nodes at batch 256 (VERDICT round-1 item 4)
        if (
            ggnn_out.is_cuda
            and ggnn_out.dtype == torch.bfloat16
            and self.label_style == "graph"
        ):
            from ..ops.flowgnn import gate_pool, mlp3

            fe = feat_embed.to(ggnn_out.dtype)
            out = gate_pool(ggnn_out, fe, self.pooling.gate_nn, graph)
            if self.encoder_mode:
                return out
            if (self.out_dim == 256 and len(self.output_layer) == 5
                    and out.shape[0] <= 1024):  # mlp3_wgrad LDS bound
                return mlp3(out, self.output_layer[0], self.output_layer[2],
                            self.output_layer[4])
            return self.output_layer(out).squeeze(-1)
        out = torch.cat([ggnn_out, feat_embed.to(ggnn_out.dtype)], dim=-1)
        if self.label_style == "graph":
            out = self.pooling(graph, out)
        if self.encoder_mode:
            return out
        return self.output_layer(out).squeeze(-1)
