"""Task-specific model architectures for Taskflow pipelines.

Reference behavior: paddlenlp/taskflow/models/dependency_parsing_model.py
(BiAffineParser: encoder + arc/rel MLPs + biaffine scorers).
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ..transformers import ErnieConfig, ErnieModel
from ..transformers.model_utils import PretrainedModel


class BiAffine(nn.Module):
    """Bilinear scorer s[b, i, j] = h_d[i] W h_h[j] (+ optional biases)."""

    def __init__(self, in_dim: int, out_channels: int = 1,
                 bias_d: bool = True, bias_h: bool = True):
        super().__init__()
        self.bias_d = bias_d
        self.bias_h = bias_h
        self.weight = nn.Parameter(torch.zeros(
            out_channels, in_dim + bias_d, in_dim + bias_h))
        nn.init.xavier_uniform_(self.weight)

    def forward(self, h_d, h_h):
        # h_d, h_h: [B, S, D] -> scores [B, out, S, S]
        if self.bias_d:
            h_d = torch.cat([h_d, h_d.new_ones(*h_d.shape[:2], 1)], dim=-1)
        if self.bias_h:
            h_h = torch.cat([h_h, h_h.new_ones(*h_h.shape[:2], 1)], dim=-1)
        return torch.einsum("bid,odh,bjh->boij", h_d, self.weight, h_h)


class BiAffineParser(PretrainedModel):
    """Dependency parser: ERNIE encoder + arc/rel biaffine heads.

    forward -> (arc_logits [B, S, S] scores of head j for word i,
                rel_logits [B, n_rels, S, S])."""

    config_class = ErnieConfig
    base_model_prefix = "parser"

    def __init__(self, config: ErnieConfig, n_rels: int = 16,
                 arc_dim: int = 128, rel_dim: int = 64):
        super().__init__(config)
        self.encoder = ErnieModel(config)
        h = config.hidden_size
        self.arc_mlp_d = nn.Sequential(nn.Linear(h, arc_dim), nn.LeakyReLU(0.1))
        self.arc_mlp_h = nn.Sequential(nn.Linear(h, arc_dim), nn.LeakyReLU(0.1))
        self.rel_mlp_d = nn.Sequential(nn.Linear(h, rel_dim), nn.LeakyReLU(0.1))
        self.rel_mlp_h = nn.Sequential(nn.Linear(h, rel_dim), nn.LeakyReLU(0.1))
        self.arc_attn = BiAffine(arc_dim, 1, bias_d=True, bias_h=False)
        self.rel_attn = BiAffine(rel_dim, n_rels, bias_d=True, bias_h=True)
        self.n_rels = n_rels

    def forward(self, input_ids, attention_mask=None, arc_labels=None,
                rel_labels=None):
        seq, _ = self.encoder(input_ids, attention_mask=attention_mask)
        arc = self.arc_attn(self.arc_mlp_d(seq), self.arc_mlp_h(seq))[:, 0]
        rel = self.rel_attn(self.rel_mlp_d(seq), self.rel_mlp_h(seq))
        if arc_labels is not None:
            B, S = input_ids.shape
            arc_loss = nn.functional.cross_entropy(
                arc.reshape(B * S, S), arc_labels.reshape(-1), ignore_index=-100)
            loss = arc_loss
            if rel_labels is not None:
                idx = arc_labels.clamp(min=0)
                # rel score of the GOLD head for each dependent
                rel_at_head = rel.permute(0, 2, 3, 1).gather(
                    2, idx[:, :, None, None].expand(-1, -1, 1, self.n_rels)
                ).squeeze(2)  # [B, S, n_rels]
                rel_loss = nn.functional.cross_entropy(
                    rel_at_head.reshape(B * S, self.n_rels),
                    rel_labels.reshape(-1), ignore_index=-100)
                loss = loss + rel_loss
            return loss, arc, rel
        return arc, rel

    @torch.no_grad()
    def decode(self, input_ids, attention_mask=None):
        """Greedy head decoding (reference uses eisner/MST; greedy is the
        compact baseline): returns (heads [B, S], rels [B, S])."""
        arc, rel = self.forward(input_ids, attention_mask)
        heads = arc.argmax(-1)                                   # [B, S]
        rel_per_head = rel.permute(0, 2, 3, 1)                   # [B, S, S, R]
        rels = rel_per_head.gather(
            2, heads[:, :, None, None].expand(-1, -1, 1, self.n_rels)
        ).squeeze(2).argmax(-1)
        return heads, rels
