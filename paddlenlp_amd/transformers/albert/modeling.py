"""ALBERT model family (reference: paddlenlp/transformers/albert/modeling.py).

The two ALBERT ideas over the shared encoder core: factorized embeddings
(embedding_size << hidden_size, projected up) and CROSS-LAYER PARAMETER
SHARING — num_hidden_groups layer instances are cycled across
num_hidden_layers applications.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..encoder import (
    EncoderEmbeddings,
    EncoderLayer,
    EncoderPooler,
    LMPredictionHead,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel
from .configuration import AlbertConfig

__all__ = ["AlbertModel", "AlbertForSequenceClassification",
           "AlbertForMaskedLM"]


class AlbertPretrainedModel(PretrainedModel):
    config_class = AlbertConfig
    base_model_prefix = "albert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class AlbertModel(AlbertPretrainedModel):
    def __init__(self, config: AlbertConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.embedding_hidden_mapping = nn.Linear(
            config.embedding_size, config.hidden_size)
        # the shared parameter groups (usually 1: every layer is the same)
        self.groups = nn.ModuleList(
            [EncoderLayer(config) for _ in range(config.num_hidden_groups)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None):
        c = self.config
        x = self.embedding_hidden_mapping(
            self.embeddings(input_ids, token_type_ids, position_ids))
        layers_per_group = c.num_hidden_layers // c.num_hidden_groups
        for i in range(c.num_hidden_layers):
            x = self.groups[i // layers_per_group](x, attention_mask)
        return x, self.pooler(x)


class AlbertForSequenceClassification(AlbertPretrainedModel):
    def __init__(self, config: AlbertConfig):
        super().__init__(config)
        self.albert = AlbertModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        _, pooled = self.albert(input_ids, token_type_ids, position_ids,
                                attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class AlbertForMaskedLM(AlbertPretrainedModel):
    _tied_weights_keys = ["predictions.decoder.weight"]

    def __init__(self, config: AlbertConfig):
        super().__init__(config)
        self.albert = AlbertModel(config)
        self.predictions = LMPredictionHead(
            config, self.albert.embeddings.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.albert(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.predictions(sequence_output)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
