"""ERNIE-M multilingual family (reference: paddlenlp/transformers/ernie_m/).

ERNIE architecture without token-type embeddings (cross-lingual alignment
pretraining is a data recipe, not an architecture change); larger
multilingual vocab and 514-position table with a +2 padding offset like
XLM-R/RoBERTa.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    EncoderEmbeddings,
    EncoderPooler,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["ErnieMConfig", "ErnieMModel",
           "ErnieMForSequenceClassification", "ErnieMForTokenClassification"]


class ErnieMConfig(PretrainedConfig):
    model_type = "ernie_m"

    attribute_map = {"num_classes": "num_labels"}

    def __init__(self, vocab_size=250002, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, attention_probs_dropout_prob=0.1,
                 max_position_embeddings=514, initializer_range=0.02,
                 layer_norm_eps=1e-5, pad_token_id=1,
                 classifier_dropout=None, num_labels=2, **kwargs):
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
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.classifier_dropout = classifier_dropout
        self.num_labels = num_labels
        self.type_vocab_size = 0  # no token-type embeddings

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ErnieMPretrainedModel(PretrainedModel):
    config_class = ErnieMConfig
    base_model_prefix = "ernie_m"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieMModel(ErnieMPretrainedModel):
    def __init__(self, config: ErnieMConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(
            config, position_offset=config.pad_token_id + 1)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, position_ids=None, attention_mask=None):
        x = self.embeddings(input_ids, None, position_ids)
        sequence_output = self.encoder(x, attention_mask)
        return sequence_output, self.pooler(sequence_output)


class ErnieMForSequenceClassification(ErnieMPretrainedModel):
    def __init__(self, config: ErnieMConfig):
        super().__init__(config)
        self.ernie_m = ErnieMModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, position_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.ernie_m(input_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class ErnieMForTokenClassification(ErnieMPretrainedModel):
    def __init__(self, config: ErnieMConfig):
        super().__init__(config)
        self.ernie_m = ErnieMModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, position_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.ernie_m(input_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(seq))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
