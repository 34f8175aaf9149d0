"""Semantic-search encoders (reference:
paddlenlp/transformers/semantic_search/modeling.py).

ErnieDualEncoder — two ERNIE towers (optionally weight-shared) with an
`output_emb_size` projection, in-batch-negative contrastive training
and cosine inference (reference :53-190) — and ErnieCrossEncoder, a
single tower scoring concatenated query/passage pairs.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ernie import ErnieConfig
from ..ernie.modeling import ErnieModel

__all__ = ["ErnieDualEncoder", "ErnieCrossEncoder"]


class ErnieDualEncoder(nn.Module):
    def __init__(self, config: ErnieConfig, output_emb_size=None,
                 share_parameters=True):
        super().__init__()
        self.query_ernie = ErnieModel(config)
        self.title_ernie = (self.query_ernie if share_parameters
                            else ErnieModel(config))
        self.proj = (nn.Linear(config.hidden_size, output_emb_size)
                     if output_emb_size else None)

    def get_pooled_embedding(self, input_ids, token_type_ids=None,
                             is_query=True):
        tower = self.query_ernie if is_query else self.title_ernie
        _, pooled = tower(input_ids, token_type_ids)
        if self.proj is not None:
            pooled = self.proj(pooled)
        return F.normalize(pooled, dim=-1)

    def cosine_sim(self, q_ids, t_ids, q_tt=None, t_tt=None):
        q = self.get_pooled_embedding(q_ids, q_tt, is_query=True)
        t = self.get_pooled_embedding(t_ids, t_tt, is_query=False)
        return (q * t).sum(-1)

    def forward(self, query_input_ids, title_input_ids,
                query_token_type_ids=None, title_token_type_ids=None,
                scale: float = 20.0):
        """In-batch-negative InfoNCE (reference forward)."""
        q = self.get_pooled_embedding(query_input_ids,
                                      query_token_type_ids, True)
        t = self.get_pooled_embedding(title_input_ids,
                                      title_token_type_ids, False)
        logits = q @ t.t() * scale
        labels = torch.arange(q.shape[0], device=q.device)
        return F.cross_entropy(logits, labels), logits


class ErnieCrossEncoder(nn.Module):
    def __init__(self, config: ErnieConfig, num_labels=2):
        super().__init__()
        self.ernie = ErnieModel(config)
        self.classifier = nn.Linear(config.hidden_size, num_labels)

    def forward(self, input_ids, token_type_ids=None, labels=None):
        _, pooled = self.ernie(input_ids, token_type_ids)
        logits = self.classifier(pooled)
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
