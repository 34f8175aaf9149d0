"""RoFormer-v2 (reference: paddlenlp/transformers/roformerv2/modeling.py).

v2 deltas over RoFormer: parameter-free variance-only Norm in place of
LayerNorm (reference Norm :39-50 — no learnable scale/bias), all
linear layers bias-free by default (`use_bias`, :258-260), and the same
interleaved rotary attention.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, expand_padding_mask, init_encoder_weights
from ..model_utils import PretrainedModel
from ..roformer.modeling import _rope_interleaved

__all__ = ["RoFormerv2Config", "RoFormerv2Model",
           "RoFormerv2ForSequenceClassification"]


class RoFormerv2Config(PretrainedConfig):
    model_type = "roformerv2"

    def __init__(self, vocab_size=12000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="relu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, use_bias=False, norm_eps=1e-12,
                 initializer_range=0.02, pad_token_id=0, num_labels=2,
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
        self.use_bias = use_bias
        self.norm_eps = norm_eps
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class Norm(nn.Module):
    """x / sqrt(mean(x^2) + eps): no learnable parameters (reference :39)."""

    def __init__(self, eps=1e-12):
        super().__init__()
        self.eps = eps

    def forward(self, x):
        var = x.pow(2).mean(dim=-1, keepdim=True)
        return x * torch.rsqrt(var + self.eps)


class RoFormerv2Layer(nn.Module):
    def __init__(self, config: RoFormerv2Config):
        super().__init__()
        h = config.hidden_size
        b = config.use_bias
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.qkv_proj = nn.Linear(h, 3 * h, bias=b)
        self.out_proj = nn.Linear(h, h, bias=b)
        self.attn_norm = Norm(config.norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size, bias=b)
        self.fc_out = nn.Linear(config.intermediate_size, h, bias=b)
        self.mlp_norm = Norm(config.norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def _cos_sin(self, S, device, dtype):
        n = self.head_dim // 2
        inv = 1.0 / (10000.0 ** (torch.arange(n, device=device).float() / n))
        freqs = torch.outer(torch.arange(S, device=device).float(), inv)
        return freqs.cos().to(dtype), freqs.sin().to(dtype)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        q, k, v = q.view(shape), k.view(shape), v.view(shape)
        cos, sin = self._cos_sin(S, x.device, x.dtype)
        q = _rope_interleaved(q, cos, sin)
        k = _rope_interleaved(k, cos, sin)
        if attention_mask is None:
            out = ops.flash_attention(q, k, v, causal=False)
        else:
            out = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                attn_mask=expand_padding_mask(attention_mask, q.dtype)
            ).transpose(1, 2)
        x = self.attn_norm(x + self.out_proj(out.reshape(B, S, H)))
        return self.mlp_norm(x + self.fc_out(self.act(self.fc_in(x))))


class RoFormerv2PretrainedModel(PretrainedModel):
    config_class = RoFormerv2Config
    base_model_prefix = "roformerv2"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class RoFormerv2Model(RoFormerv2PretrainedModel):
    def __init__(self, config: RoFormerv2Config):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = Norm(config.norm_eps)
        self.layers = nn.ModuleList(
            [RoFormerv2Layer(config)
             for _ in range(config.num_hidden_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        x = self.embeddings(input_ids)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x


class RoFormerv2ForSequenceClassification(RoFormerv2PretrainedModel):
    def __init__(self, config: RoFormerv2Config):
        super().__init__(config)
        self.roformerv2 = RoFormerv2Model(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq = self.roformerv2(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(torch.tanh(seq[:, 0])))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
