"""PPMiniLM family (reference: paddlenlp/transformers/ppminilm/modeling.py).

6-layer Chinese MiniLM student with the BERT architecture (relu default in
the reference configs is actually gelu for ppminilm-6l-768h); pure rebadge
over the shared encoder core with sequence-classification and QA heads.
"""
from __future__ import annotations

import torch.nn as nn

from ..bert.configuration import BertConfig
from ..encoder import (
    EncoderEmbeddings,
    EncoderPooler,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["PPMiniLMConfig", "PPMiniLMModel",
           "PPMiniLMForSequenceClassification", "PPMiniLMForQuestionAnswering"]


class PPMiniLMConfig(BertConfig):
    model_type = "ppminilm"


class PPMiniLMPretrainedModel(PretrainedModel):
    config_class = PPMiniLMConfig
    base_model_prefix = "ppminilm"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class PPMiniLMModel(PPMiniLMPretrainedModel):
    def __init__(self, config: PPMiniLMConfig):
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


class PPMiniLMForSequenceClassification(PPMiniLMPretrainedModel):
    def __init__(self, config: PPMiniLMConfig):
        super().__init__(config)
        self.ppminilm = PPMiniLMModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.ppminilm(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return nn.functional.cross_entropy(logits, labels.view(-1)), logits
        return logits


class PPMiniLMForQuestionAnswering(PPMiniLMPretrainedModel):
    def __init__(self, config: PPMiniLMConfig):
        super().__init__(config)
        self.ppminilm = PPMiniLMModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, _ = self.ppminilm(input_ids, token_type_ids, attention_mask)
        start, end = self.classifier(seq).chunk(2, dim=-1)
        return start.squeeze(-1), end.squeeze(-1)
