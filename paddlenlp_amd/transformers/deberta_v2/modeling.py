"""DeBERTa-v2 (reference: paddlenlp/transformers/deberta_v2/modeling.py).

v2 deltas over DeBERTa: LOG-BUCKET relative positions
(make_log_bucket_position — near positions exact, far positions in
log-spaced buckets up to `position_buckets`), a LayerNorm over the
shared relative-embedding table before projection, and an optional
depthwise token-conv branch after the first layer (conv_kernel_size).
The disentangled c2c + c2p + p2c score itself matches v1.
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

__all__ = ["DebertaV2Config", "DebertaV2Model",
           "DebertaV2ForSequenceClassification", "DebertaV2ForMaskedLM"]


class DebertaV2Config(PretrainedConfig):
    model_type = "deberta-v2"

    attribute_map = {"num_classes": "num_labels"}

    def __init__(self, vocab_size=128100, hidden_size=1536,
                 num_hidden_layers=24, num_attention_heads=24,
                 intermediate_size=6144, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 max_relative_positions=512, position_buckets=256,
                 conv_kernel_size=0, conv_act="gelu",
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
        self.max_position_embeddings = max_position_embeddings
        self.max_relative_positions = max_relative_positions
        self.position_buckets = position_buckets
        self.conv_kernel_size = conv_kernel_size
        self.conv_act = conv_act
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.classifier_dropout = classifier_dropout
        self.num_labels = num_labels
        self.type_vocab_size = 0

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


def make_log_bucket_position(rel: torch.Tensor, buckets: int,
                             max_pos: int) -> torch.Tensor:
    """v2 log buckets: |rel| < buckets/2 exact; beyond, log-spaced up to
    max_pos (reference deberta_v2 make_log_bucket_position)."""
    sign = torch.sign(rel)
    mid = buckets // 2
    abs_pos = torch.where((rel < mid) & (rel > -mid),
                          torch.full_like(rel, mid - 1), rel.abs())
    log_pos = torch.ceil(
        torch.log(abs_pos.float() / mid)
        / math.log((max_pos - 1) / mid) * (mid - 1)) + mid
    return torch.where(abs_pos <= mid, rel.float(),
                       log_pos * sign).long()


class DisentangledSelfAttentionV2(nn.Module):
    def __init__(self, config: DebertaV2Config):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.qkv_proj = nn.Linear(h, 3 * h, bias=True)
        self.out_proj = nn.Linear(h, h, bias=True)
        self.buckets = config.position_buckets
        self.max_pos = config.max_relative_positions
        # v2 shares ONE projection pair for the shared rel table
        self.pos_key_proj = nn.Linear(h, h, bias=True)
        self.pos_query_proj = nn.Linear(h, h, bias=True)

    def _heads(self, t, B, S):
        return t.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)

    def _bucket(self, S, device):
        pos = torch.arange(S, device=device)
        rel = pos[None, :] - pos[:, None]
        k = self.buckets
        return make_log_bucket_position(rel, k, self.max_pos).clamp(
            -k, k - 1) + k

    def forward(self, x, rel_embeddings, attention_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q = self._heads(q, B, S)
        k = self._heads(k, B, S)
        v = self._heads(v, B, S)

        idx = self._bucket(S, x.device)                         # [S,S]
        pk = self.pos_key_proj(rel_embeddings).view(
            -1, self.num_heads, self.head_dim)
        pq = self.pos_query_proj(rel_embeddings).view(
            -1, self.num_heads, self.head_dim)

        scale = 1.0 / math.sqrt(self.head_dim * 3)
        c2c = q @ k.transpose(-1, -2)
        c2p = torch.einsum("bhid,khd->bhik", q, pk).gather(
            -1, idx[None, None].expand(B, self.num_heads, S, S))
        p2c = torch.einsum("bhjd,khd->bhjk", k, pq).gather(
            -1, idx.t()[None, None].expand(B, self.num_heads, S, S)
        ).transpose(-1, -2)
        scores = (c2c + c2p + p2c) * scale
        if attention_mask is not None:
            scores = scores + expand_padding_mask(attention_mask, scores.dtype)
        out = scores.softmax(-1) @ v
        return self.out_proj(out.transpose(1, 2).reshape(B, S, H))


class DebertaV2Layer(nn.Module):
    def __init__(self, config: DebertaV2Config):
        super().__init__()
        h = config.hidden_size
        self.self_attn = DisentangledSelfAttentionV2(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, rel_embeddings, attention_mask=None):
        x = self.attn_norm(x + self.self_attn(x, rel_embeddings,
                                              attention_mask))
        return self.mlp_norm(x + self.fc_out(self.act(self.fc_in(x))))


class ConvLayer(nn.Module):
    """v2's token-conv branch merged after the first layer's output."""

    def __init__(self, config: DebertaV2Config):
        super().__init__()
        h = config.hidden_size
        ks = config.conv_kernel_size
        self.conv = nn.Conv1d(h, h, ks, padding=(ks - 1) // 2, groups=1)
        self.norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.conv_act]

    def forward(self, x, residual):
        y = self.act(self.conv(x.transpose(1, 2)).transpose(1, 2))
        return self.norm(residual + y)


class DebertaV2PretrainedModel(PretrainedModel):
    config_class = DebertaV2Config
    base_model_prefix = "deberta"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class DebertaV2Model(DebertaV2PretrainedModel):
    def __init__(self, config: DebertaV2Config):
        super().__init__(config)
        self.embeddings = nn.Embedding(config.vocab_size, config.hidden_size,
                                       padding_idx=config.pad_token_id)
        self.embed_norm = nn.LayerNorm(config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.rel_embeddings = nn.Embedding(
            2 * config.position_buckets, config.hidden_size)
        # v2: the shared rel table is LayerNormed before projection
        self.rel_norm = nn.LayerNorm(config.hidden_size,
                                     eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [DebertaV2Layer(config)
             for _ in range(config.num_hidden_layers)])
        self.conv = (ConvLayer(config)
                     if config.conv_kernel_size > 0 else None)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, attention_mask=None):
        x = self.embed_norm(self.embeddings(input_ids))
        rel = self.rel_norm(self.rel_embeddings.weight)
        for i, layer in enumerate(self.layers):
            y = layer(x, rel, attention_mask)
            if i == 0 and self.conv is not None:
                y = self.conv(x, y)
            x = y
        return x


class DebertaV2ForSequenceClassification(DebertaV2PretrainedModel):
    def __init__(self, config: DebertaV2Config):
        super().__init__(config)
        self.deberta = DebertaV2Model(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.pooler = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)
        self.init_weights()

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq = self.deberta(input_ids, attention_mask)
        pooled = torch.tanh(self.pooler(seq[:, 0]))
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class DebertaV2ForMaskedLM(DebertaV2PretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: DebertaV2Config):
        super().__init__(config)
        self.deberta = DebertaV2Model(config)
        self.cls = LMPredictionHead(config, self.deberta.embeddings.weight)
        self.init_weights()

    def forward(self, input_ids, attention_mask=None, labels=None):
        logits = self.cls(self.deberta(input_ids, attention_mask))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
