"""ERNIE-Gram family (reference: paddlenlp/transformers/ernie_gram/).

Explicit n-gram masked LM pretraining gives the checkpoints their value; the
architecture at fine-tune time is the BERT/ERNIE encoder (the optional
rel_pos path is off in every released config), so the tower is the shared
post-LN encoder core with the standard task heads.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import (
    EncoderEmbeddings,
    EncoderPooler,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["ErnieGramConfig", "ErnieGramModel",
           "ErnieGramForSequenceClassification",
           "ErnieGramForTokenClassification", "ErnieGramForQuestionAnswering"]


class ErnieGramConfig(BertConfig):
    model_type = "ernie_gram"

    def __init__(self, vocab_size=18018, **kwargs):
        kwargs.setdefault("hidden_act", "gelu")
        super().__init__(vocab_size=vocab_size, **kwargs)


class ErnieGramPretrainedModel(PretrainedModel):
    config_class = ErnieGramConfig
    base_model_prefix = "ernie_gram"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieGramModel(ErnieGramPretrainedModel):
    def __init__(self, config: ErnieGramConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        x = self.embeddings(input_ids, token_type_ids)
        seq = self.encoder(x, attention_mask)
        return seq, self.pooler(seq)


class ErnieGramForSequenceClassification(ErnieGramPretrainedModel):
    def __init__(self, config: ErnieGramConfig):
        super().__init__(config)
        self.ernie_gram = ErnieGramModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.ernie_gram(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class ErnieGramForTokenClassification(ErnieGramPretrainedModel):
    def __init__(self, config: ErnieGramConfig):
        super().__init__(config)
        self.ernie_gram = ErnieGramModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.ernie_gram(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(seq))
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits


class ErnieGramForQuestionAnswering(ErnieGramPretrainedModel):
    def __init__(self, config: ErnieGramConfig):
        super().__init__(config)
        self.ernie_gram = ErnieGramModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, _ = self.ernie_gram(input_ids, token_type_ids, attention_mask)
        start, end = self.classifier(seq).chunk(2, dim=-1)
        return start.squeeze(-1), end.squeeze(-1)
