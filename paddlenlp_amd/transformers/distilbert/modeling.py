"""DistilBERT family (reference: paddlenlp/transformers/distilbert/).

BERT architecture distilled: no token-type embeddings, no pooler (the
sequence-classification head pools CLS through a pre-classifier ReLU
projection).  Built on the shared encoder core.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    EncoderEmbeddings,
    LMPredictionHead,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["DistilBertConfig", "DistilBertModel",
           "DistilBertForSequenceClassification", "DistilBertForMaskedLM"]


class DistilBertConfig(PretrainedConfig):
    model_type = "distilbert"

    attribute_map = {"num_classes": "num_labels", "dim": "hidden_size",
                     "n_layers": "num_hidden_layers",
                     "n_heads": "num_attention_heads"}

    def __init__(self, vocab_size=30522, hidden_size=768,
                 num_hidden_layers=6, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, attention_probs_dropout_prob=0.1,
                 max_position_embeddings=512, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0,
                 classifier_dropout=0.2, num_labels=2, **kwargs):
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
        self.type_vocab_size = 0  # distilled away

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class DistilBertPretrainedModel(PretrainedModel):
    config_class = DistilBertConfig
    base_model_prefix = "distilbert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class DistilBertModel(DistilBertPretrainedModel):
    def __init__(self, config: DistilBertConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.encoder = TransformerEncoder(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, attention_mask=None):
        x = self.embeddings(input_ids)
        return self.encoder(x, attention_mask)


class DistilBertForSequenceClassification(DistilBertPretrainedModel):
    def __init__(self, config: DistilBertConfig):
        super().__init__(config)
        self.distilbert = DistilBertModel(config)
        self.pre_classifier = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout = nn.Dropout(config.classifier_dropout)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq = self.distilbert(input_ids, attention_mask)
        pooled = F.relu(self.pre_classifier(seq[:, 0]))
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class DistilBertForMaskedLM(DistilBertPretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: DistilBertConfig):
        super().__init__(config)
        self.distilbert = DistilBertModel(config)
        self.cls = LMPredictionHead(
            config, self.distilbert.embeddings.word_embeddings.weight)

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq = self.distilbert(input_ids, attention_mask)
        logits = self.cls(seq)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
