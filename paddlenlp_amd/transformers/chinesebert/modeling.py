"""ChineseBERT (reference: paddlenlp/transformers/chinesebert/modeling.py).

BERT whose input embedding fuses three views of each character:
word-piece embedding, a PINYIN embedding (Conv1d over the romanization
id sequence of each char, max-pooled), and a GLYPH embedding (per-char
visual features projected by `glyph_map`) — concatenated and fused by
`map_fc` back to the hidden size.  The glyph table ships as pretrained
font-bitmap weights in the reference checkpoint; it is a learned
[vocab, glyph_embedding_dim] table here, loaded from the checkpoint the
same way.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    EncoderLayer,
    EncoderPooler,
    LMPredictionHead,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["ChineseBertConfig", "ChineseBertModel",
           "ChineseBertForSequenceClassification"]


class ChineseBertConfig(PretrainedConfig):
    model_type = "chinesebert"

    def __init__(self, vocab_size=23236, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, pinyin_map_size=32,
                 pinyin_embedding_size=128, pinyin_locs=8,
                 glyph_embedding_dim=1728, initializer_range=0.02,
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
        self.attention_probs_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.pinyin_map_size = pinyin_map_size
        self.pinyin_embedding_size = pinyin_embedding_size
        self.pinyin_locs = pinyin_locs
        self.glyph_embedding_dim = glyph_embedding_dim
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class PinyinEmbedding(nn.Module):
    """Conv over each char's pinyin-letter ids, max-pooled (reference
    PinyinEmbedding)."""

    def __init__(self, config: ChineseBertConfig):
        super().__init__()
        self.embedding = nn.Embedding(config.pinyin_map_size,
                                      config.pinyin_embedding_size)
        self.conv = nn.Conv1d(config.pinyin_embedding_size,
                              config.hidden_size, kernel_size=2)

    def forward(self, pinyin_ids):
        # [B, S, locs] -> [B*S, locs, E] -> conv -> maxpool -> [B, S, H]
        B, S, L = pinyin_ids.shape
        e = self.embedding(pinyin_ids.reshape(B * S, L))
        y = self.conv(e.transpose(1, 2))
        y = F.max_pool1d(y, y.shape[-1]).squeeze(-1)
        return y.view(B, S, -1)


class FusionEmbedding(nn.Module):
    def __init__(self, config: ChineseBertConfig):
        super().__init__()
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h,
                                            padding_idx=config.pad_token_id)
        self.pinyin_embeddings = PinyinEmbedding(config)
        self.glyph_embeddings = nn.Embedding(config.vocab_size,
                                             config.glyph_embedding_dim)
        self.glyph_map = nn.Linear(config.glyph_embedding_dim, h)
        self.map_fc = nn.Linear(3 * h, h)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, input_ids, pinyin_ids=None, token_type_ids=None):
        B, S = input_ids.shape
        word = self.word_embeddings(input_ids)
        if pinyin_ids is None:
            pinyin_ids = torch.zeros(B, S, 8, dtype=torch.long,
                                     device=input_ids.device)
        pinyin = self.pinyin_embeddings(pinyin_ids)
        glyph = self.glyph_map(self.glyph_embeddings(input_ids))
        x = self.map_fc(torch.cat([word, pinyin, glyph], dim=-1))
        pos = torch.arange(S, device=input_ids.device)
        x = x + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        return self.dropout(self.norm(x))


class ChineseBertPretrainedModel(PretrainedModel):
    config_class = ChineseBertConfig
    base_model_prefix = "chinesebert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ChineseBertModel(ChineseBertPretrainedModel):
    def __init__(self, config: ChineseBertConfig):
        super().__init__(config)
        self.embeddings = FusionEmbedding(config)
        self.layers = nn.ModuleList(
            [EncoderLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, pinyin_ids=None, token_type_ids=None,
                attention_mask=None):
        x = self.embeddings(input_ids, pinyin_ids, token_type_ids)
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x, self.pooler(x)


class ChineseBertForSequenceClassification(ChineseBertPretrainedModel):
    def __init__(self, config: ChineseBertConfig):
        super().__init__(config)
        self.chinesebert = ChineseBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, pinyin_ids=None, token_type_ids=None,
                attention_mask=None, labels=None):
        _, pooled = self.chinesebert(input_ids, pinyin_ids, token_type_ids,
                                     attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
