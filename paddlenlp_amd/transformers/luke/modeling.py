"""LUKE (reference: paddlenlp/transformers/luke/modeling.py).

Entity-aware RoBERTa: a separate entity-embedding stream (small
`entity_emb_size` table projected up, multi-position span embeddings,
reference EntityEmbeddings :171-206) is concatenated with the word
stream, and self-attention uses FOUR query matrices — w2w, w2e, e2w,
e2e — one per (query-kind, key-kind) pair, over shared keys/values
(reference LukeSelfAttention).  Heads: entity classification (typing)
and entity-span classification.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["LukeConfig", "LukeModel", "LukeForEntityClassification"]


class LukeConfig(PretrainedConfig):
    model_type = "luke"

    def __init__(self, vocab_size=50267, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=514,
                 type_vocab_size=1, entity_vocab_size=500000,
                 entity_emb_size=256, entity_pad_id=0,
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
        self.entity_vocab_size = entity_vocab_size
        self.entity_emb_size = entity_emb_size
        self.entity_pad_id = entity_pad_id
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class EntityEmbeddings(nn.Module):
    """Small entity table projected up + mean of span-position embeddings
    (reference :171-206)."""

    def __init__(self, config: LukeConfig):
        super().__init__()
        self.entity_embeddings = nn.Embedding(
            config.entity_vocab_size, config.entity_emb_size, padding_idx=0)
        self.entity_embedding_dense = (
            nn.Linear(config.entity_emb_size, config.hidden_size, bias=False)
            if config.entity_emb_size != config.hidden_size else None)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, config.hidden_size)
        self.norm = nn.LayerNorm(config.hidden_size,
                                 eps=config.layer_norm_eps)

    def forward(self, entity_ids, entity_position_ids):
        e = self.entity_embeddings(entity_ids)
        if self.entity_embedding_dense is not None:
            e = self.entity_embedding_dense(e)
        # entity_position_ids: [B, E, span_len], -1 padded
        mask = (entity_position_ids >= 0).to(e.dtype)
        pos = self.position_embeddings(entity_position_ids.clamp(min=0))
        pos = (pos * mask.unsqueeze(-1)).sum(2) / \
            mask.sum(2).clamp(min=1).unsqueeze(-1)
        return self.norm(e + pos)


class EntityAwareAttention(nn.Module):
    """Four query projections, shared k/v (the LUKE mechanism)."""

    def __init__(self, config: LukeConfig):
        super().__init__()
        h = config.hidden_size
        self.nh = config.num_attention_heads
        self.dh = config.head_dim
        self.query = nn.Linear(h, h)          # w2w
        self.w2e_query = nn.Linear(h, h)
        self.e2w_query = nn.Linear(h, h)
        self.e2e_query = nn.Linear(h, h)
        self.key = nn.Linear(h, h)
        self.value = nn.Linear(h, h)
        self.out = nn.Linear(h, h)

    def _h(self, t):
        B, S, _ = t.shape
        return t.view(B, S, self.nh, self.dh).transpose(1, 2)

    def forward(self, words, entities, word_mask=None, entity_mask=None):
        B, Sw, H = words.shape
        Se = entities.shape[1]
        k = self._h(self.key(torch.cat([words, entities], dim=1)))
        v = self._h(self.value(torch.cat([words, entities], dim=1)))
        # per-pair query projections
        qw2w = self._h(self.query(words))
        qw2e = self._h(self.w2e_query(words))
        qe2w = self._h(self.e2w_query(entities))
        qe2e = self._h(self.e2e_query(entities))
        kw, ke = k[:, :, :Sw], k[:, :, Sw:]
        s_w = torch.cat([qw2w @ kw.transpose(-1, -2),
                         qw2e @ ke.transpose(-1, -2)], dim=-1)
        s_e = torch.cat([qe2w @ kw.transpose(-1, -2),
                         qe2e @ ke.transpose(-1, -2)], dim=-1)
        scores = torch.cat([s_w, s_e], dim=2) / math.sqrt(self.dh)
        if word_mask is not None or entity_mask is not None:
            wm = word_mask if word_mask is not None else \
                torch.ones(B, Sw, device=words.device)
            em = entity_mask if entity_mask is not None else \
                torch.ones(B, Se, device=words.device)
            keymask = torch.cat([wm, em], dim=1).to(scores.dtype)
            scores = scores + (1.0 - keymask)[:, None, None, :] * \
                torch.finfo(scores.dtype).min
        out = scores.softmax(-1) @ v
        out = self.out(out.transpose(1, 2).reshape(B, Sw + Se, H))
        return out[:, :Sw], out[:, Sw:]


class LukeLayer(nn.Module):
    def __init__(self, config: LukeConfig):
        super().__init__()
        h = config.hidden_size
        self.attn = EntityAwareAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.ff_in = nn.Linear(h, config.intermediate_size)
        self.ff_out = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, words, entities, word_mask=None, entity_mask=None):
        aw, ae = self.attn(words, entities, word_mask, entity_mask)
        x = self.attn_norm(torch.cat([words, entities], dim=1) +
                           torch.cat([aw, ae], dim=1))
        x = self.ff_norm(x + self.ff_out(self.act(self.ff_in(x))))
        Sw = words.shape[1]
        return x[:, :Sw], x[:, Sw:]


class LukePretrainedModel(PretrainedModel):
    config_class = LukeConfig
    base_model_prefix = "luke"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class LukeModel(LukePretrainedModel):
    def __init__(self, config: LukeConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.entity_embeddings = EntityEmbeddings(config)
        self.layers = nn.ModuleList(
            [LukeLayer(config) for _ in range(config.num_hidden_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, entity_ids=None, entity_position_ids=None,
                attention_mask=None, entity_attention_mask=None):
        B, S = input_ids.shape
        pos = torch.arange(S, device=input_ids.device)
        w = self.embed_norm(self.embeddings(input_ids) +
                            self.position_embeddings(pos))
        if entity_ids is None:
            entity_ids = torch.zeros(B, 1, dtype=torch.long,
                                     device=input_ids.device)
            entity_position_ids = torch.zeros(B, 1, 1, dtype=torch.long,
                                              device=input_ids.device)
        e = self.entity_embeddings(entity_ids, entity_position_ids)
        for layer in self.layers:
            w, e = layer(w, e, attention_mask, entity_attention_mask)
        return w, e


class LukeForEntityClassification(LukePretrainedModel):
    """Entity typing: classify the first entity's final representation."""

    def __init__(self, config: LukeConfig):
        super().__init__(config)
        self.luke = LukeModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, entity_ids=None, entity_position_ids=None,
                attention_mask=None, entity_attention_mask=None, labels=None):
        _, e = self.luke(input_ids, entity_ids, entity_position_ids,
                         attention_mask, entity_attention_mask)
        logits = self.classifier(self.dropout(e[:, 0]))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
