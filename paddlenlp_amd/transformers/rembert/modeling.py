"""RemBERT family (reference: paddlenlp/transformers/rembert/modeling.py).

Rebalanced-embedding multilingual BERT: SMALL decoupled input embeddings
(input_embedding_size, e.g. 256) projected up to hidden_size before the
tower, and an UNTIED output-embedding MLM head of its own
(output_embedding_size).  The tower itself is the shared post-LN encoder.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["RemBertConfig", "RemBertModel",
           "RemBertForSequenceClassification", "RemBertForMaskedLM"]


class RemBertConfig(BertConfig):
    model_type = "rembert"

    def __init__(self, input_embedding_size=256, output_embedding_size=1664,
                 **kwargs):
        super().__init__(**kwargs)
        self.input_embedding_size = input_embedding_size
        self.output_embedding_size = output_embedding_size


class RemBertPretrainedModel(PretrainedModel):
    config_class = RemBertConfig
    base_model_prefix = "rembert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class RemBertModel(RemBertPretrainedModel):
    def __init__(self, config: RemBertConfig):
        super().__init__(config)
        e = config.input_embedding_size
        self.word_embeddings = nn.Embedding(config.vocab_size, e,
                                            padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings, e)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, e)
        self.embed_norm = nn.LayerNorm(e, eps=config.layer_norm_eps)
        self.embed_dropout = nn.Dropout(config.hidden_dropout_prob)
        self.embedding_hidden_mapping_in = nn.Linear(e, config.hidden_size)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        S = input_ids.shape[1]
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        pos = torch.arange(S, device=input_ids.device)
        x = (self.word_embeddings(input_ids)
             + self.position_embeddings(pos)
             + self.token_type_embeddings(token_type_ids))
        x = self.embed_dropout(self.embed_norm(x))
        x = self.embedding_hidden_mapping_in(x)
        seq = self.encoder(x, attention_mask)
        return seq, self.pooler(seq)


class RemBertForSequenceClassification(RemBertPretrainedModel):
    def __init__(self, config: RemBertConfig):
        super().__init__(config)
        self.rembert = RemBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.rembert(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class RemBertForMaskedLM(RemBertPretrainedModel):
    """Untied output embeddings: hidden -> output_embedding_size -> vocab."""

    def __init__(self, config: RemBertConfig):
        super().__init__(config)
        self.rembert = RemBertModel(config)
        self.dense = nn.Linear(config.hidden_size, config.output_embedding_size)
        self.act = ACT2FN[config.hidden_act]
        self.norm = nn.LayerNorm(config.output_embedding_size,
                                 eps=config.layer_norm_eps)
        self.decoder = nn.Linear(config.output_embedding_size, config.vocab_size)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.rembert(input_ids, token_type_ids, attention_mask)
        logits = self.decoder(self.norm(self.act(self.dense(seq))))
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits
