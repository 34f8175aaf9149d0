"""GAU-alpha (reference: paddlenlp/transformers/gau_alpha/modeling.py).

Gated Attention Unit encoder: each layer is a single GAU — shared
u/v gates (swish) around a low-rank attention whose q/k come from ONE
`attention_key_size`-wide projection via per-dim scale+offset
(reference ScaleOffset :121-122), RoPE on q/k, and relu² attention
normalization (:141-146) instead of softmax.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["GAUAlphaConfig", "GAUAlphaModel",
           "GAUAlphaForSequenceClassification"]


class GAUAlphaConfig(PretrainedConfig):
    model_type = "gau_alpha"

    def __init__(self, vocab_size=12000, hidden_size=768,
                 intermediate_size=1536, num_hidden_layers=24,
                 attention_key_size=128, hidden_act="swish",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, use_bias=False,
                 normalization="softmax_plus", attention_scale=True,
                 initializer_range=0.02, layer_norm_eps=1e-12,
                 pad_token_id=0, num_labels=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.num_hidden_layers = num_hidden_layers
        self.attention_key_size = attention_key_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.use_bias = use_bias
        self.normalization = normalization
        self.attention_scale = attention_scale
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels


class ScaleOffset(nn.Module):
    """Per-dim learned scale (+optional offset), reference :121-122."""

    def __init__(self, size, offset=True):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(size))
        self.bias = nn.Parameter(torch.zeros(size)) if offset else None

    def forward(self, x):
        x = x * self.weight
        if self.bias is not None:
            x = x + self.bias
        return x


def _rope(x, cos, sin):
    x0, x1 = x[..., 0::2], x[..., 1::2]
    return torch.stack([x0 * cos - x1 * sin, x1 * cos + x0 * sin],
                       dim=-1).flatten(-2)


def attention_normalize(a, mask=None, method="softmax_plus"):
    """reference :146: softmax_plus scales logits by log(n)/log(512)."""
    if method == "softmax_plus":
        n = a.shape[-1] if mask is None else \
            mask.sum(dim=-1, keepdim=True).clamp(min=1).unsqueeze(1)
        ln = torch.log(torch.as_tensor(n, dtype=a.dtype, device=a.device)) \
            if not torch.is_tensor(n) else torch.log(n.to(a.dtype))
        a = a * ln / torch.log(torch.tensor(512.0, device=a.device))
    if mask is not None:
        a = a + (1.0 - mask.to(a.dtype)).unsqueeze(1) * torch.finfo(a.dtype).min
    return F.softmax(a, dim=-1)


class GAULayer(nn.Module):
    def __init__(self, config: GAUAlphaConfig):
        super().__init__()
        h, e, s = (config.hidden_size, config.intermediate_size,
                   config.attention_key_size)
        self.e, self.s = e, s
        self.uv_dense = nn.Linear(h, 2 * e + s, bias=config.use_bias)
        self.o_dense = nn.Linear(e, h, bias=config.use_bias)
        self.q_scaleoffset = ScaleOffset(s, offset=config.use_bias)
        self.k_scaleoffset = ScaleOffset(s, offset=config.use_bias)
        self.norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.attention_scale = config.attention_scale
        self.normalization = config.normalization
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, attention_mask=None, rope=None):
        uv = self.uv_dense(self.norm(x))
        u, v, z = torch.split(uv, [self.e, self.e, self.s], dim=-1)
        u, v = F.silu(u), F.silu(v)
        q = self.q_scaleoffset(z)
        k = self.k_scaleoffset(z)
        if rope is not None:
            cos, sin = rope
            q, k = _rope(q, cos, sin), _rope(k, cos, sin)
        a = q @ k.transpose(-1, -2)                 # [B,S,S]
        if self.attention_scale:
            a = a / self.s ** 0.5
        A = attention_normalize(a.unsqueeze(1), attention_mask,
                                self.normalization).squeeze(1)
        return x + self.dropout(self.o_dense(u * (A @ v)))


class GAUAlphaPretrainedModel(PretrainedModel):
    config_class = GAUAlphaConfig
    base_model_prefix = "gau_alpha"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class GAUAlphaModel(GAUAlphaPretrainedModel):
    def __init__(self, config: GAUAlphaConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [GAULayer(config) for _ in range(config.num_hidden_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def _rope_tables(self, S, device, dtype):
        s = self.config.attention_key_size
        inv = 1.0 / (10000 ** (torch.arange(0, s, 2, device=device).float() / s))
        freqs = torch.outer(torch.arange(S, device=device).float(), inv)
        return freqs.cos().to(dtype), freqs.sin().to(dtype)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        S = input_ids.shape[1]
        x = self.embeddings(input_ids)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        rope = self._rope_tables(S, x.device, x.dtype)
        for layer in self.layers:
            x = layer(x, attention_mask, rope)
        return x


class GAUAlphaForSequenceClassification(GAUAlphaPretrainedModel):
    def __init__(self, config: GAUAlphaConfig):
        super().__init__(config)
        self.gau_alpha = GAUAlphaModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq = self.gau_alpha(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(seq[:, 0]))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
