"""ERNIE-CTM (reference: paddlenlp/transformers/ernie_ctm/modeling.py).

Chinese text mining encoder: `cls_num` parallel [CLS] slots whose
token-type/position ids are zeroed (reference ErnieCtmEmbeddings
:86-127), a content-summary pooler over cls slot 0, and the WordtagTask
head (token classification over the content region + sentence-level
cls logits).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import EncoderLayer, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ErnieCtmConfig", "ErnieCtmModel", "ErnieCtmWordtagModel"]


class ErnieCtmConfig(PretrainedConfig):
    model_type = "ernie_ctm"

    def __init__(self, vocab_size=23000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, cls_num=2, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, num_labels=2,
                 num_tag=265, **kwargs):
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
        self.cls_num = cls_num
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels
        self.num_tag = num_tag

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ErnieCtmPretrainedModel(PretrainedModel):
    config_class = ErnieCtmConfig
    base_model_prefix = "ernie_ctm"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieCtmModel(ErnieCtmPretrainedModel):
    def __init__(self, config: ErnieCtmConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [EncoderLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = nn.Linear(h, h)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        B, S = input_ids.shape
        cls_num = self.config.cls_num
        # the first cls_num slots carry position/type id 0 (reference
        # :98-110): content positions start after them
        content_len = S - cls_num
        pos = torch.cat([
            torch.zeros(cls_num, dtype=torch.long,
                        device=input_ids.device),
            torch.arange(content_len, device=input_ids.device)])
        x = self.embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is not None:
            tt = token_type_ids.clone()
            tt[:, :cls_num] = 0
            x = x + self.token_type_embeddings(tt)
        x = self.embed_norm(x)
        for layer in self.layers:
            x = layer(x, attention_mask)
        pooled = torch.tanh(self.pooler(x[:, 0]))
        return x, pooled


class ErnieCtmWordtagModel(ErnieCtmPretrainedModel):
    """Wordtag: token tags over the content region + cls sentence logits."""

    def __init__(self, config: ErnieCtmConfig):
        super().__init__(config)
        self.ernie_ctm = ErnieCtmModel(config)
        self.tag_classifier = nn.Linear(config.hidden_size, config.num_tag)
        self.sent_classifier = nn.Linear(config.hidden_size,
                                         config.num_labels)
        self.init_weights()

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                tag_labels=None):
        seq, pooled = self.ernie_ctm(input_ids, token_type_ids,
                                     attention_mask)
        tag_logits = self.tag_classifier(seq[:, self.config.cls_num:])
        sent_logits = self.sent_classifier(pooled)
        if tag_labels is not None:
            loss = F.cross_entropy(
                tag_logits.reshape(-1, self.config.num_tag),
                tag_labels.reshape(-1), ignore_index=-100)
            return loss, tag_logits, sent_logits
        return tag_logits, sent_logits
