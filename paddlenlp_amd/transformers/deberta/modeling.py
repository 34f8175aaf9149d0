"""DeBERTa family (reference: paddlenlp/transformers/deberta/modeling.py).

Disentangled attention: the score decomposes into content-content plus
content-to-position and position-to-content terms computed against a shared
relative-position embedding table (log-free v1 buckets clipped at
max_relative_positions); absolute positions never enter the stream.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    LMPredictionHead,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["DebertaConfig", "DebertaModel",
           "DebertaForSequenceClassification", "DebertaForMaskedLM"]


class DebertaConfig(PretrainedConfig):
    model_type = "deberta"

    attribute_map = {"num_classes": "num_labels"}

    def __init__(self, vocab_size=50265, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, attention_probs_dropout_prob=0.1,
                 max_position_embeddings=512, max_relative_positions=128,
                 initializer_range=0.02, layer_norm_eps=1e-7,
                 pad_token_id=0, classifier_dropout=None, num_labels=2,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = attention_probs_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.max_relative_positions = max_relative_positions
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.classifier_dropout = classifier_dropout
        self.num_labels = num_labels
        self.type_vocab_size = 0

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class DisentangledSelfAttention(nn.Module):
    """score = c2c + c2p + p2c over a clipped relative-position table."""

    def __init__(self, config: DebertaConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.qkv_proj = nn.Linear(h, 3 * h, bias=True)
        self.out_proj = nn.Linear(h, h, bias=True)
        self.k_span = config.max_relative_positions
        self.pos_key_proj = nn.Linear(h, h, bias=True)
        self.pos_query_proj = nn.Linear(h, h, bias=True)

    def _heads(self, t, B, S):
        return t.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)

    def forward(self, x, rel_embeddings, attention_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q = self._heads(q, B, S)
        k = self._heads(k, B, S)
        v = self._heads(v, B, S)

        # relative index matrix clipped to [-k, k-1] -> [0, 2k-1]
        pos = torch.arange(S, device=x.device)
        rel = (pos[None, :] - pos[:, None]).clamp(-self.k_span,
                                                  self.k_span - 1) + self.k_span
        # project the rel table into key/query spaces: [2k, H] -> heads
        pk = self.pos_key_proj(rel_embeddings).view(
            -1, self.num_heads, self.head_dim)      # [2k, h, d]
        pq = self.pos_query_proj(rel_embeddings).view(
            -1, self.num_heads, self.head_dim)

        scale = 1.0 / math.sqrt(self.head_dim * 3)  # 3 score terms
        c2c = q @ k.transpose(-1, -2)                          # [B,h,S,S]
        # c2p: q_i . pk[rel(i,j)]
        c2p_all = torch.einsum("bhid,khd->bhik", q, pk)        # [B,h,S,2k]
        c2p = c2p_all.gather(
            -1, rel[None, None].expand(B, self.num_heads, S, S))
        # p2c: k_j . pq[rel(j,i)]
        p2c_all = torch.einsum("bhjd,khd->bhjk", k, pq)        # [B,h,S,2k]
        relT = (pos[None, :] - pos[:, None]).t().clamp(
            -self.k_span, self.k_span - 1) + self.k_span       # rel(j,i)
        p2c = p2c_all.gather(
            -1, relT[None, None].expand(B, self.num_heads, S, S)).transpose(-1, -2)
        scores = (c2c + c2p + p2c) * scale
        if attention_mask is not None:
            scores = scores + expand_padding_mask(attention_mask, scores.dtype)
        probs = scores.softmax(-1)
        out = probs @ v
        return self.out_proj(out.transpose(1, 2).reshape(B, S, H))


class DebertaLayer(nn.Module):
    def __init__(self, config: DebertaConfig):
        super().__init__()
        h = config.hidden_size
        self.self_attn = DisentangledSelfAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, rel_embeddings, attention_mask=None):
        x = self.attn_norm(x + self.self_attn(x, rel_embeddings, attention_mask))
        return self.mlp_norm(x + self.fc_out(self.act(self.fc_in(x))))


class DebertaPretrainedModel(PretrainedModel):
    config_class = DebertaConfig
    base_model_prefix = "deberta"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class DebertaModel(DebertaPretrainedModel):
    def __init__(self, config: DebertaConfig):
        super().__init__(config)
        self.embeddings = nn.Embedding(config.vocab_size, config.hidden_size,
                                       padding_idx=config.pad_token_id)
        self.embed_norm = nn.LayerNorm(config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.rel_embeddings = nn.Embedding(
            2 * config.max_relative_positions, config.hidden_size)
        self.layers = nn.ModuleList(
            [DebertaLayer(config) for _ in range(config.num_hidden_layers)])

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, attention_mask=None):
        x = self.embed_norm(self.embeddings(input_ids))
        rel = self.rel_embeddings.weight
        for layer in self.layers:
            x = layer(x, rel, attention_mask)
        return x


class DebertaForSequenceClassification(DebertaPretrainedModel):
    def __init__(self, config: DebertaConfig):
        super().__init__(config)
        self.deberta = DebertaModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq = self.deberta(input_ids, attention_mask)
        pooled = torch.tanh(self.pooler(seq[:, 0]))
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class DebertaForMaskedLM(DebertaPretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: DebertaConfig):
        super().__init__(config)
        self.deberta = DebertaModel(config)
        self.cls = LMPredictionHead(config, self.deberta.embeddings.weight)

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq = self.deberta(input_ids, attention_mask)
        logits = self.cls(seq)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
