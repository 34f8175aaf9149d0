"""NeZha family (reference: paddlenlp/transformers/nezha/modeling.py).

Huawei NeZha: BERT shape but NO learned absolute positions — attention uses
a fixed SINUSOIDAL relative-position table rel[i, j, d] (distances clipped
to ±max_relative_position), contributing to both the scores
(q_i · rel[i,j]) and the context (sum_j p_ij rel[i,j]).  The table is a
pure function of positions, generated on the fly (meta-device safe), and
the relative terms are einsum contractions — the whole attention stays one
fused matmul chain on MFMA.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["NeZhaConfig", "NeZhaModel", "NeZhaForSequenceClassification",
           "NeZhaForTokenClassification", "NeZhaForQuestionAnswering"]


class NeZhaConfig(BertConfig):
    model_type = "nezha"

    def __init__(self, max_relative_position: int = 64,
                 use_relative_position: bool = True, **kwargs):
        super().__init__(**kwargs)
        self.max_relative_position = max_relative_position
        self.use_relative_position = use_relative_position


def relative_position_table(length: int, depth: int, max_rel: int,
                            device, dtype) -> torch.Tensor:
    """rel[i, j, :] = sinusoidal(clip(j - i, ±max_rel) + max_rel).

    Reference nezha/modeling.py:81 builds this via one-hot matmul; here it's
    a direct gather from the sinusoid table."""
    vocab = 2 * max_rel + 1
    pos = torch.arange(vocab, dtype=torch.float32, device=device)[:, None]
    i = torch.arange(depth, dtype=torch.float32, device=device)[None, :]
    angle = pos / torch.pow(10000.0, (i // 2) * 2 / depth)
    table = torch.where((torch.arange(depth, device=device) % 2) == 0,
                        angle.sin(), angle.cos())
    dist = (torch.arange(length, device=device)[None, :]
            - torch.arange(length, device=device)[:, None]).clamp(-max_rel, max_rel)
    return table[(dist + max_rel)].to(dtype)  # [S, S, depth]


class NeZhaAttention(nn.Module):
    def __init__(self, config: NeZhaConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = h // self.num_heads
        self.max_rel = config.max_relative_position
        self.qkv_proj = nn.Linear(h, 3 * h)
        self.out_proj = nn.Linear(h, h)
        self.dropout = nn.Dropout(config.attention_probs_dropout_prob)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        k = k.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        v = v.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        rel = relative_position_table(S, self.head_dim, self.max_rel,
                                      x.device, torch.float32)
        scores = q @ k.transpose(-1, -2)
        scores = scores + torch.einsum("bhid,ijd->bhij", q.float(), rel).to(q.dtype)
        scores = scores / math.sqrt(self.head_dim)
        if attention_mask is not None:
            scores = scores + expand_padding_mask(attention_mask, scores.dtype)
        probs = self.dropout(F.softmax(scores, dim=-1))
        ctx = probs @ v
        ctx = ctx + torch.einsum("bhij,ijd->bhid", probs.float(), rel).to(v.dtype)
        ctx = ctx.transpose(1, 2).reshape(B, S, H)
        return self.out_proj(ctx)


class NeZhaLayer(nn.Module):
    def __init__(self, config: NeZhaConfig):
        super().__init__()
        h = config.hidden_size
        self.attention = NeZhaAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, attention_mask=None):
        x = self.attn_norm(x + self.dropout(self.attention(x, attention_mask)))
        mlp = self.fc_out(self.act(self.fc_in(x)))
        return self.mlp_norm(x + self.dropout(mlp))


class NeZhaPretrainedModel(PretrainedModel):
    config_class = NeZhaConfig
    base_model_prefix = "nezha"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class NeZhaModel(NeZhaPretrainedModel):
    def __init__(self, config: NeZhaConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size,
                                            padding_idx=config.pad_token_id)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size,
                                                  config.hidden_size)
        self.embed_norm = nn.LayerNorm(config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.embed_dropout = nn.Dropout(config.hidden_dropout_prob)
        self.layers = nn.ModuleList(
            [NeZhaLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        # no absolute positions: relative attention carries all order info
        x = self.word_embeddings(input_ids) \
            + self.token_type_embeddings(token_type_ids)
        x = self.embed_dropout(self.embed_norm(x))
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x, self.pooler(x)


class NeZhaForSequenceClassification(NeZhaPretrainedModel):
    def __init__(self, config: NeZhaConfig):
        super().__init__(config)
        self.nezha = NeZhaModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.nezha(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class NeZhaForTokenClassification(NeZhaPretrainedModel):
    def __init__(self, config: NeZhaConfig):
        super().__init__(config)
        self.nezha = NeZhaModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.nezha(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(seq))
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits


class NeZhaForQuestionAnswering(NeZhaPretrainedModel):
    def __init__(self, config: NeZhaConfig):
        super().__init__(config)
        self.nezha = NeZhaModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, _ = self.nezha(input_ids, token_type_ids, attention_mask)
        start, end = self.classifier(seq).chunk(2, dim=-1)
        return start.squeeze(-1), end.squeeze(-1)
