"""SqueezeBERT (reference: paddlenlp/transformers/squeezebert/modeling.py).

BERT with every position-wise linear replaced by a grouped 1-D
convolution over the [B, C, S] layout (q/k/v/attention-output with
`q_groups`.. and the FFN pair with `intermediate_groups`/`output_groups`)
— the SqueezeBERT efficiency trick.  Attention math itself is standard.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["SqueezeBertConfig", "SqueezeBertModel",
           "SqueezeBertForSequenceClassification"]


class SqueezeBertConfig(PretrainedConfig):
    model_type = "squeezebert"

    def __init__(self, vocab_size=30528, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, q_groups=4, k_groups=4, v_groups=4,
                 post_attention_groups=1, intermediate_groups=4,
                 output_groups=4, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, num_labels=2,
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
        self.type_vocab_size = type_vocab_size
        self.q_groups = q_groups
        self.k_groups = k_groups
        self.v_groups = v_groups
        self.post_attention_groups = post_attention_groups
        self.intermediate_groups = intermediate_groups
        self.output_groups = output_groups
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels


class ConvDropoutLayerNorm(nn.Module):
    """conv1x1(groups) + residual + LayerNorm over channels."""

    def __init__(self, cin, cout, groups, config):
        super().__init__()
        self.conv = nn.Conv1d(cin, cout, 1, groups=groups)
        self.norm = nn.LayerNorm(cout, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, residual):
        # x: [B, C, S] -> norm over channel dim in [B, S, C]
        y = self.dropout(self.conv(x)).transpose(1, 2)
        return self.norm(y + residual.transpose(1, 2)).transpose(1, 2)


class SqueezeBertLayer(nn.Module):
    def __init__(self, config: SqueezeBertConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = h // config.num_attention_heads
        self.q = nn.Conv1d(h, h, 1, groups=config.q_groups)
        self.k = nn.Conv1d(h, h, 1, groups=config.k_groups)
        self.v = nn.Conv1d(h, h, 1, groups=config.v_groups)
        self.post_attn = ConvDropoutLayerNorm(
            h, h, config.post_attention_groups, config)
        self.ffn_in = nn.Conv1d(h, config.intermediate_size, 1,
                                groups=config.intermediate_groups)
        self.ffn_out = ConvDropoutLayerNorm(
            config.intermediate_size, h, config.output_groups, config)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, attention_mask=None):
        # x: [B, C, S] throughout (the SqueezeBERT data layout)
        B, C, S = x.shape
        shp = (B, self.num_heads, self.head_dim, S)
        q = self.q(x).view(shp).transpose(-1, -2)   # [B,nh,S,hd]
        k = self.k(x).view(shp).transpose(-1, -2)
        v = self.v(x).view(shp).transpose(-1, -2)
        mask = (expand_padding_mask(attention_mask, q.dtype)
                if attention_mask is not None else None)
        a = F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
        a = a.transpose(-1, -2).reshape(B, C, S)
        x = self.post_attn(a, x)
        y = self.act(self.ffn_in(x))
        return self.ffn_out(y, x)


class SqueezeBertPretrainedModel(PretrainedModel):
    config_class = SqueezeBertConfig
    base_model_prefix = "squeezebert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class SqueezeBertModel(SqueezeBertPretrainedModel):
    def __init__(self, config: SqueezeBertConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [SqueezeBertLayer(config)
             for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x).transpose(1, 2)      # -> [B, C, S]
        for layer in self.layers:
            x = layer(x, attention_mask)
        x = x.transpose(1, 2)
        return x, self.pooler(x)


class SqueezeBertForSequenceClassification(SqueezeBertPretrainedModel):
    def __init__(self, config: SqueezeBertConfig):
        super().__init__(config)
        self.squeezebert = SqueezeBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.squeezebert(input_ids, token_type_ids,
                                     attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
