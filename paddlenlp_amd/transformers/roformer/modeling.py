"""RoFormer family (reference: paddlenlp/transformers/roformer/modeling.py).

BERT-style bidirectional encoder whose self-attention applies ROTARY
position embedding to q/k (interleaved pairs over the full head dim) — no
absolute position table.  Attention runs through the flash seam when
unmasked.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    LMPredictionHead,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["RoFormerConfig", "RoFormerModel",
           "RoFormerForSequenceClassification", "RoFormerForMaskedLM"]


class RoFormerConfig(PretrainedConfig):
    model_type = "roformer"

    attribute_map = {"num_classes": "num_labels"}

    def __init__(self, vocab_size=50000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, attention_probs_dropout_prob=0.1,
                 max_position_embeddings=1536, rotary_value=False,
                 initializer_range=0.02, layer_norm_eps=1e-12,
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
        self.rotary_value = rotary_value
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.classifier_dropout = classifier_dropout
        self.num_labels = num_labels
        self.type_vocab_size = 0

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


def _rope_interleaved(x, cos, sin):
    """x [B, S, H, D]; interleaved-pair rotation over the full head dim."""
    x0 = x[..., 0::2]
    x1 = x[..., 1::2]
    c = cos[None, :, None, :]
    s = sin[None, :, None, :]
    return torch.stack([x0 * c - x1 * s, x1 * c + x0 * s], dim=-1).flatten(-2)


class RoFormerAttention(nn.Module):
    def __init__(self, config: RoFormerConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.qkv_proj = nn.Linear(h, 3 * h, bias=True)
        self.out_proj = nn.Linear(h, h, bias=True)
        self.rotary_value = config.rotary_value

    def _cos_sin(self, S, device, dtype):
        n = self.head_dim // 2
        inv = 1.0 / (10000.0 ** (torch.arange(n, device=device).float() / n))
        freqs = torch.outer(torch.arange(S, device=device).float(), inv)
        return freqs.cos().to(dtype), freqs.sin().to(dtype)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        q = q.view(shape)
        k = k.view(shape)
        v = v.view(shape)
        cos, sin = self._cos_sin(S, x.device, x.dtype)
        q = _rope_interleaved(q, cos, sin)
        k = _rope_interleaved(k, cos, sin)
        if self.rotary_value:
            v = _rope_interleaved(v, cos, sin)
        if attention_mask is None:
            out = ops.flash_attention(q, k, v, causal=False)
        else:
            add_mask = expand_padding_mask(attention_mask, q.dtype)
            out = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                attn_mask=add_mask).transpose(1, 2)
        return self.out_proj(out.reshape(B, S, H))


class RoFormerLayer(nn.Module):
    def __init__(self, config: RoFormerConfig):
        super().__init__()
        h = config.hidden_size
        self.self_attn = RoFormerAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, attention_mask=None):
        x = self.attn_norm(x + self.self_attn(x, attention_mask))
        return self.mlp_norm(x + self.fc_out(self.act(self.fc_in(x))))


class RoFormerPretrainedModel(PretrainedModel):
    config_class = RoFormerConfig
    base_model_prefix = "roformer"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class RoFormerModel(RoFormerPretrainedModel):
    def __init__(self, config: RoFormerConfig):
        super().__init__(config)
        self.embeddings = nn.Embedding(config.vocab_size, config.hidden_size,
                                       padding_idx=config.pad_token_id)
        self.embed_norm = nn.LayerNorm(config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [RoFormerLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, attention_mask=None):
        x = self.embed_norm(self.embeddings(input_ids))
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x, self.pooler(x)


class RoFormerForSequenceClassification(RoFormerPretrainedModel):
    def __init__(self, config: RoFormerConfig):
        super().__init__(config)
        self.roformer = RoFormerModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, attention_mask=None, labels=None):
        _, pooled = self.roformer(input_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class RoFormerForMaskedLM(RoFormerPretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: RoFormerConfig):
        super().__init__(config)
        self.roformer = RoFormerModel(config)
        self.cls = LMPredictionHead(config, self.roformer.embeddings.weight)

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq, _ = self.roformer(input_ids, attention_mask)
        logits = self.cls(seq)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
