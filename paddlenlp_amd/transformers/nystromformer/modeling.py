"""Nystromformer (reference: paddlenlp/transformers/nystromformer/modeling.py).

Self-attention approximated through `num_landmarks` segment-mean
landmarks: softmax(Q K~^T) · pinv(softmax(Q~ K~^T)) · softmax(Q~ K^T) V,
with the Moore-Penrose pseudo-inverse computed by the reference's
6-step Newton-Schulz iteration (its `iterative_inv`), plus the depthwise
conv residual on V (`conv_kernel_size`).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    LMPredictionHead,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["NystromformerConfig", "NystromformerModel",
           "NystromformerForSequenceClassification"]


class NystromformerConfig(PretrainedConfig):
    model_type = "nystromformer"

    def __init__(self, vocab_size=30000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu_new",
                 hidden_dropout_prob=0.1, max_position_embeddings=510,
                 type_vocab_size=2, num_landmarks=64, segment_means_seq_len=64,
                 conv_kernel_size=65, inv_coeff_init_option=False,
                 initializer_range=0.02, layer_norm_eps=1e-5,
                 pad_token_id=1, num_labels=2, **kwargs):
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
        self.num_landmarks = num_landmarks
        self.segment_means_seq_len = segment_means_seq_len
        self.conv_kernel_size = conv_kernel_size
        self.inv_coeff_init_option = inv_coeff_init_option
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


def iterative_inv(mat: torch.Tensor, n_iter: int = 6) -> torch.Tensor:
    """Newton-Schulz pseudo-inverse (reference iterative_inv)."""
    I = torch.eye(mat.shape[-1], device=mat.device, dtype=mat.dtype)
    z = mat.transpose(-1, -2) / (mat.abs().sum(dim=-2).max(dim=-1, keepdim=True)
                                 .values.unsqueeze(-1) *
                                 mat.abs().sum(dim=-1).max(dim=-1, keepdim=True)
                                 .values.unsqueeze(-1))
    for _ in range(n_iter):
        kv = mat @ z
        z = 0.25 * z @ (13 * I - kv @ (15 * I - kv @ (7 * I - kv)))
    return z


class NystromAttention(nn.Module):
    def __init__(self, config: NystromformerConfig):
        super().__init__()
        h = config.hidden_size
        self.nh = config.num_attention_heads
        self.dh = config.head_dim
        self.m = config.num_landmarks
        self.query = nn.Linear(h, h)
        self.key = nn.Linear(h, h)
        self.value = nn.Linear(h, h)
        self.out = nn.Linear(h, h)
        ks = config.conv_kernel_size
        self.conv = None
        if ks:
            self.conv = nn.Conv2d(self.nh, self.nh, (ks, 1),
                                  padding=(ks // 2, 0), groups=self.nh,
                                  bias=False)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        shp = (B, S, self.nh, self.dh)
        scale = 1.0 / math.sqrt(math.sqrt(self.dh))
        q = self.query(x).view(shp).transpose(1, 2) * scale
        k = self.key(x).view(shp).transpose(1, 2) * scale
        v = self.value(x).view(shp).transpose(1, 2)
        bias = None
        if attention_mask is not None:
            bias = (1.0 - attention_mask.to(q.dtype)) * torch.finfo(q.dtype).min

        if S <= self.m or S % self.m != 0:
            score = q @ k.transpose(-1, -2)
            if bias is not None:
                score = score + bias[:, None, None, :]
            o = F.softmax(score, dim=-1) @ v
        else:
            seg = S // self.m
            ql = q.reshape(B, self.nh, self.m, seg, self.dh).mean(dim=-2)
            kl = k.reshape(B, self.nh, self.m, seg, self.dh).mean(dim=-2)
            k1 = q @ kl.transpose(-1, -2)                       # [B,h,S,m]
            k2 = ql @ kl.transpose(-1, -2)                      # [B,h,m,m]
            k3 = ql @ k.transpose(-1, -2)                       # [B,h,m,S]
            if bias is not None:
                k3 = k3 + bias[:, None, None, :]
            k1 = F.softmax(k1, dim=-1)
            k2 = F.softmax(k2, dim=-1)
            k3 = F.softmax(k3, dim=-1)
            o = k1 @ (iterative_inv(k2) @ (k3 @ v))
        if self.conv is not None:
            o = o + self.conv(v)
        return self.out(o.transpose(1, 2).reshape(B, S, H))


class NystromformerLayer(nn.Module):
    def __init__(self, config: NystromformerConfig):
        super().__init__()
        h = config.hidden_size
        self.attn = NystromAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, attention_mask=None):
        x = self.attn_norm(x + self.dropout(self.attn(x, attention_mask)))
        y = self.fc_out(self.act(self.fc_in(x)))
        return self.mlp_norm(x + self.dropout(y))


class NystromformerPretrainedModel(PretrainedModel):
    config_class = NystromformerConfig
    base_model_prefix = "nystromformer"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class NystromformerModel(NystromformerPretrainedModel):
    def __init__(self, config: NystromformerConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList([
            NystromformerLayer(config)
            for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        x = self.embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x, self.pooler(x)


class NystromformerForSequenceClassification(NystromformerPretrainedModel):
    def __init__(self, config: NystromformerConfig):
        super().__init__(config)
        self.nystromformer = NystromformerModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.nystromformer(input_ids, token_type_ids,
                                       attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
