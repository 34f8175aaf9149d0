"""MPNet family (reference: paddlenlp/transformers/mpnet/modeling.py).

Masked-and-permuted encoder: BERT shape + learned absolute positions PLUS a
T5-style bucketed relative-attention BIAS shared by all layers (one
Embedding(num_buckets -> num_heads) owned by the encoder).  The bias is an
additive attention mask, so every layer still runs through one SDPA call.
Reuses the T5 bucket function (t5/modeling.py:41 in this repo).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    LMPredictionHead,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel
from ..t5.modeling import relative_position_bucket

__all__ = ["MPNetConfig", "MPNetModel", "MPNetForSequenceClassification",
           "MPNetForMaskedLM"]


class MPNetConfig(BertConfig):
    model_type = "mpnet"

    def __init__(self, relative_attention_num_buckets: int = 32,
                 pad_token_id: int = 1, **kwargs):
        kwargs.setdefault("vocab_size", 30527)
        super().__init__(pad_token_id=pad_token_id, **kwargs)
        self.relative_attention_num_buckets = relative_attention_num_buckets


class MPNetLayer(nn.Module):
    def __init__(self, config: MPNetConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = h // self.num_heads
        self.qkv_proj = nn.Linear(h, 3 * h)
        self.out_proj = nn.Linear(h, h)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.attn_dropout_p = config.attention_probs_dropout_prob

    def forward(self, x, bias, attention_mask=None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        k = k.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        v = v.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        mask = bias
        if attention_mask is not None:
            mask = mask + expand_padding_mask(attention_mask, x.dtype)
        attn = F.scaled_dot_product_attention(
            q, k, v, attn_mask=mask,
            dropout_p=self.attn_dropout_p if self.training else 0.0)
        x = self.attn_norm(x + self.dropout(
            self.out_proj(attn.transpose(1, 2).reshape(B, S, H))))
        mlp = self.fc_out(self.act(self.fc_in(x)))
        return self.mlp_norm(x + self.dropout(mlp))


class MPNetPretrainedModel(PretrainedModel):
    config_class = MPNetConfig
    base_model_prefix = "mpnet"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class MPNetModel(MPNetPretrainedModel):
    def __init__(self, config: MPNetConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size,
                                            padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings,
                                                config.hidden_size)
        self.embed_norm = nn.LayerNorm(config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.embed_dropout = nn.Dropout(config.hidden_dropout_prob)
        self.relative_attention_bias = nn.Embedding(
            config.relative_attention_num_buckets, config.num_attention_heads)
        self.layers = nn.ModuleList(
            [MPNetLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.word_embeddings

    def compute_position_bias(self, S, device):
        ctx = torch.arange(S, device=device)[:, None]
        mem = torch.arange(S, device=device)[None, :]
        buckets = relative_position_bucket(
            mem - ctx, bidirectional=True,
            num_buckets=self.config.relative_attention_num_buckets,
            max_distance=128)
        return self.relative_attention_bias(buckets).permute(2, 0, 1)[None]

    def forward(self, input_ids, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.word_embeddings(input_ids) + self.position_embeddings(pos)
        x = self.embed_dropout(self.embed_norm(x))
        bias = self.compute_position_bias(S, input_ids.device).to(x.dtype)
        for layer in self.layers:
            x = layer(x, bias, attention_mask)
        return x, self.pooler(x)


class MPNetForSequenceClassification(MPNetPretrainedModel):
    def __init__(self, config: MPNetConfig):
        super().__init__(config)
        self.mpnet = MPNetModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, attention_mask=None, labels=None):
        _, pooled = self.mpnet(input_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class MPNetForMaskedLM(MPNetPretrainedModel):
    def __init__(self, config: MPNetConfig):
        super().__init__(config)
        self.mpnet = MPNetModel(config)
        self.lm_head = LMPredictionHead(
            config, embedding_weights=self.mpnet.word_embeddings.weight)

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq, _ = self.mpnet(input_ids, attention_mask)
        logits = self.lm_head(seq)
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits
