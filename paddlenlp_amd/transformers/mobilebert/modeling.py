"""MobileBERT (reference: paddlenlp/transformers/mobilebert/modeling.py).

Bottleneck transformer: 128-wide trigram-fused embeddings projected to
the 512 body width, per-layer input/output bottlenecks squeezing
attention + FFN compute to `intra_bottleneck_size`, a stack of
`num_feedforward_networks` FFNs per layer, and NoNorm (elementwise
affine, no statistics) in place of LayerNorm.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    LMPredictionHead,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["MobileBertConfig", "MobileBertModel",
           "MobileBertForSequenceClassification"]


class MobileBertConfig(PretrainedConfig):
    model_type = "mobilebert"

    def __init__(self, vocab_size=30522, hidden_size=512,
                 embedding_size=128, num_hidden_layers=24,
                 num_attention_heads=4, intermediate_size=512,
                 intra_bottleneck_size=128, num_feedforward_networks=4,
                 hidden_act="relu", hidden_dropout_prob=0.0,
                 max_position_embeddings=512, type_vocab_size=2,
                 normalization_type="no_norm", initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, num_labels=2,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.embedding_size = embedding_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.intra_bottleneck_size = intra_bottleneck_size
        self.num_feedforward_networks = num_feedforward_networks
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.normalization_type = normalization_type
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels


class NoNorm(nn.Module):
    """Elementwise affine without statistics (the MobileBERT latency trick)."""

    def __init__(self, size):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(size))
        self.bias = nn.Parameter(torch.zeros(size))

    def forward(self, x):
        return x * self.weight + self.bias


def _norm(config, size):
    if config.normalization_type == "no_norm":
        return NoNorm(size)
    return nn.LayerNorm(size, eps=config.layer_norm_eps)


class MobileBertLayer(nn.Module):
    def __init__(self, config: MobileBertConfig):
        super().__init__()
        h, b = config.hidden_size, config.intra_bottleneck_size
        self.bottleneck_in = nn.Linear(h, b)
        self.bottleneck_in_norm = _norm(config, b)
        self.num_heads = config.num_attention_heads
        self.head_dim = b // config.num_attention_heads
        self.query = nn.Linear(b, b)
        self.key = nn.Linear(b, b)
        self.value = nn.Linear(b, b)
        self.attn_out = nn.Linear(b, b)
        self.attn_norm = _norm(config, b)
        self.ffns = nn.ModuleList()
        for _ in range(config.num_feedforward_networks):
            self.ffns.append(nn.ModuleList([
                nn.Linear(b, config.intermediate_size),
                nn.Linear(config.intermediate_size, b),
                _norm(config, b)]))
        self.bottleneck_out = nn.Linear(b, h)
        self.bottleneck_out_norm = _norm(config, h)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        z = self.bottleneck_in_norm(self.bottleneck_in(x))
        shp = (B, S, self.num_heads, self.head_dim)
        q = self.query(z).view(shp).transpose(1, 2)
        k = self.key(z).view(shp).transpose(1, 2)
        v = self.value(z).view(shp).transpose(1, 2)
        mask = (expand_padding_mask(attention_mask, z.dtype)
                if attention_mask is not None else None)
        a = F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
        a = a.transpose(1, 2).reshape(B, S, -1)
        z = self.attn_norm(z + self.attn_out(a))
        for fc_in, fc_out, norm in self.ffns:
            z = norm(z + fc_out(self.act(fc_in(z))))
        return self.bottleneck_out_norm(x + self.dropout(
            self.bottleneck_out(z)))


class MobileBertPretrainedModel(PretrainedModel):
    config_class = MobileBertConfig
    base_model_prefix = "mobilebert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class MobileBertModel(MobileBertPretrainedModel):
    def __init__(self, config: MobileBertConfig):
        super().__init__(config)
        e, h = config.embedding_size, config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, e,
                                            padding_idx=config.pad_token_id)
        # trigram fusion: concat[x_{t-1}, x_t, x_{t+1}] -> hidden
        self.embedding_transform = nn.Linear(3 * e, h)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = _norm(config, h)
        self.layers = nn.ModuleList(
            [MobileBertLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        B, S = input_ids.shape
        w = self.word_embeddings(input_ids)
        tri = torch.cat([F.pad(w, (0, 0, 1, 0))[:, :-1], w,
                         F.pad(w, (0, 0, 0, 1))[:, 1:]], dim=-1)
        x = self.embedding_transform(tri)
        pos = torch.arange(S, device=input_ids.device)
        x = x + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x, self.pooler(x)


class MobileBertForSequenceClassification(MobileBertPretrainedModel):
    def __init__(self, config: MobileBertConfig):
        super().__init__(config)
        self.mobilebert = MobileBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.mobilebert(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
