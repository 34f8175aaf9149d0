"""Funnel Transformer (reference: paddlenlp/transformers/funnel/modeling.py).

Hourglass encoder: `block_sizes` blocks whose hidden sequence is
mean-pooled 2x at each block boundary — the first layer of a block runs
pool-query attention (q from the pooled stream, k/v from the unpooled
one) — plus the decoder that upsamples the final coarse stream back to
full length, adds the first-block residual, and runs 2 refinement
layers for token-level tasks.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["FunnelConfig", "FunnelModel",
           "FunnelForSequenceClassification", "FunnelForTokenClassification"]


class FunnelConfig(PretrainedConfig):
    model_type = "funnel"

    def __init__(self, vocab_size=30522, hidden_size=768,
                 num_attention_heads=12, intermediate_size=3072,
                 block_sizes=(4, 4, 4), num_decoder_layers=2,
                 hidden_act="gelu", hidden_dropout_prob=0.1,
                 max_position_embeddings=512, type_vocab_size=3,
                 pool_q_only=True, initializer_range=0.02,
                 layer_norm_eps=1e-9, pad_token_id=0, num_labels=2,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.block_sizes = list(block_sizes)
        self.num_hidden_layers = sum(self.block_sizes)
        self.num_decoder_layers = num_decoder_layers
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.pool_q_only = pool_q_only
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class FunnelAttention(nn.Module):
    """Cross-length attention: q may come from a pooled stream."""

    def __init__(self, config: FunnelConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.q_proj = nn.Linear(h, h)
        self.k_proj = nn.Linear(h, h)
        self.v_proj = nn.Linear(h, h)
        self.out_proj = nn.Linear(h, h)

    def forward(self, q_in, kv_in, key_mask=None):
        B, Sq, H = q_in.shape
        Sk = kv_in.shape[1]
        q = self.q_proj(q_in).view(B, Sq, self.num_heads,
                                   self.head_dim).transpose(1, 2)
        k = self.k_proj(kv_in).view(B, Sk, self.num_heads,
                                    self.head_dim).transpose(1, 2)
        v = self.v_proj(kv_in).view(B, Sk, self.num_heads,
                                    self.head_dim).transpose(1, 2)
        mask = None
        if key_mask is not None:
            mask = expand_padding_mask(key_mask, q.dtype)
        out = F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
        return self.out_proj(out.transpose(1, 2).reshape(B, Sq, H))


class FunnelLayer(nn.Module):
    def __init__(self, config: FunnelConfig):
        super().__init__()
        h = config.hidden_size
        self.attn = FunnelAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, q_in, kv_in, key_mask=None):
        x = self.attn_norm(q_in + self.dropout(self.attn(q_in, kv_in,
                                                         key_mask)))
        return self.mlp_norm(x + self.dropout(
            self.fc_out(self.act(self.fc_in(x)))))


def _pool(x, mask=None):
    """Mean-pool stride 2 along the sequence (reference pooling)."""
    S = x.shape[1]
    if S % 2:
        x = F.pad(x, (0, 0, 0, 1))
        if mask is not None:
            mask = F.pad(mask, (0, 1))
    x = x.reshape(x.shape[0], -1, 2, x.shape[-1]).mean(dim=2)
    if mask is not None:
        mask = mask.reshape(mask.shape[0], -1, 2).amax(dim=2)
    return x, mask


class FunnelPretrainedModel(PretrainedModel):
    config_class = FunnelConfig
    base_model_prefix = "funnel"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class FunnelModel(FunnelPretrainedModel):
    def __init__(self, config: FunnelConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.blocks = nn.ModuleList()
        for bs in config.block_sizes:
            self.blocks.append(nn.ModuleList(
                [FunnelLayer(config) for _ in range(bs)]))
        self.decoder_layers = nn.ModuleList(
            [FunnelLayer(config) for _ in range(config.num_decoder_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def encode(self, input_ids, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.embed_norm(self.embeddings(input_ids) +
                            self.position_embeddings(pos))
        mask = attention_mask
        first_block_out = None
        for bi, block in enumerate(self.blocks):
            for li, layer in enumerate(block):
                if bi > 0 and li == 0:
                    # pooled query attends to the UNPOOLED stream
                    pooled, pooled_mask = _pool(x, mask)
                    x = layer(pooled, x, mask)
                    mask = pooled_mask
                else:
                    x = layer(x, x, mask)
            if bi == 0:
                first_block_out = x
        return x, first_block_out, mask

    def forward(self, input_ids, attention_mask=None):
        coarse, first, _ = self.encode(input_ids, attention_mask)
        # decoder: nearest-neighbour upsample back to full length,
        # residual with the first block's full-resolution stream
        S = first.shape[1]
        factor = 2 ** (len(self.config.block_sizes) - 1)
        up = coarse.repeat_interleave(factor, dim=1)[:, :S]
        x = up + first
        for layer in self.decoder_layers:
            x = layer(x, x, attention_mask)
        return x, coarse


class FunnelForSequenceClassification(FunnelPretrainedModel):
    """Classification reads the COARSE stream's first token (no decoder
    pass needed — the funnel's efficiency win for sentence tasks)."""

    def __init__(self, config: FunnelConfig):
        super().__init__(config)
        self.funnel = FunnelModel(config)
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)
        self.init_weights()

    def forward(self, input_ids, attention_mask=None, labels=None):
        coarse, _, _ = self.funnel.encode(input_ids, attention_mask)
        pooled = torch.tanh(self.dense(coarse[:, 0]))
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class FunnelForTokenClassification(FunnelPretrainedModel):
    def __init__(self, config: FunnelConfig):
        super().__init__(config)
        self.funnel = FunnelModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)
        self.init_weights()

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq, _ = self.funnel(input_ids, attention_mask)
        logits = self.classifier(self.dropout(seq))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
