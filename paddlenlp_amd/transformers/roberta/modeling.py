"""RoBERTa model family (reference: paddlenlp/transformers/roberta/modeling.py).

BERT architecture with RoBERTa specifics: positions offset past the padding
index (pad_token_id + 1), no NSP objective, a two-layer tanh classification
head over <s>, and an MLM head (dense + GELU + LN + tied decoder).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..encoder import (
    EncoderEmbeddings,
    EncoderPooler,
    LMPredictionHead,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel
from .configuration import RobertaConfig

__all__ = [
    "RobertaModel",
    "RobertaPretrainedModel",
    "RobertaForSequenceClassification",
    "RobertaForTokenClassification",
    "RobertaForQuestionAnswering",
    "RobertaForMaskedLM",
]


class RobertaPretrainedModel(PretrainedModel):
    config_class = RobertaConfig
    base_model_prefix = "roberta"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class RobertaModel(RobertaPretrainedModel):
    def __init__(self, config: RobertaConfig):
        super().__init__(config)
        # RoBERTa positions start right after the padding index
        self.embeddings = EncoderEmbeddings(
            config, position_offset=config.pad_token_id + 1)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None):
        x = self.embeddings(input_ids, token_type_ids, position_ids)
        sequence_output = self.encoder(x, attention_mask)
        return sequence_output, self.pooler(sequence_output)


class RobertaClassificationHead(nn.Module):
    """dense + tanh + dense over the <s> hidden state."""

    def __init__(self, config):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.out_proj = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, sequence_output):
        x = self.dropout(sequence_output[:, 0])
        x = torch.tanh(self.dense(x))
        return self.out_proj(self.dropout(x))


class RobertaForSequenceClassification(RobertaPretrainedModel):
    def __init__(self, config: RobertaConfig):
        super().__init__(config)
        self.roberta = RobertaModel(config)
        self.classifier = RobertaClassificationHead(config)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.roberta(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(sequence_output)
        if labels is not None:
            if self.config.num_labels == 1:
                loss = F.mse_loss(logits.squeeze(-1), labels.float())
            else:
                loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class RobertaForTokenClassification(RobertaPretrainedModel):
    def __init__(self, config: RobertaConfig):
        super().__init__(config)
        self.roberta = RobertaModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.roberta(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(sequence_output))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class RobertaForQuestionAnswering(RobertaPretrainedModel):
    def __init__(self, config: RobertaConfig):
        super().__init__(config)
        self.roberta = RobertaModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, start_positions=None, end_positions=None):
        sequence_output, _ = self.roberta(
            input_ids, token_type_ids, position_ids, attention_mask)
        start_logits, end_logits = self.classifier(sequence_output).unbind(-1)
        if start_positions is not None and end_positions is not None:
            S = start_logits.shape[1]
            loss = 0.5 * (
                F.cross_entropy(start_logits, start_positions.clamp(0, S - 1))
                + F.cross_entropy(end_logits, end_positions.clamp(0, S - 1)))
            return loss, start_logits, end_logits
        return start_logits, end_logits


class RobertaForMaskedLM(RobertaPretrainedModel):
    _tied_weights_keys = ["lm_head.decoder.weight"]

    def __init__(self, config: RobertaConfig):
        super().__init__(config)
        self.roberta = RobertaModel(config)
        self.lm_head = LMPredictionHead(
            config, self.roberta.embeddings.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.roberta(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.lm_head(sequence_output)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
