"""BERT model family (reference: paddlenlp/transformers/bert/modeling.py).

Post-LN bidirectional encoder built on the shared
``paddlenlp_amd.transformers.encoder`` core: BertModel + the task heads the
reference ships (sequence/token classification, QA, multiple choice,
masked-LM, MLM+NSP pretraining with BertPretrainingCriterion).
"""
from __future__ import annotations

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
from .configuration import BertConfig

__all__ = [
    "BertModel",
    "BertPretrainedModel",
    "BertForSequenceClassification",
    "BertForTokenClassification",
    "BertForQuestionAnswering",
    "BertForMultipleChoice",
    "BertForMaskedLM",
    "BertForPretraining",
    "BertPretrainingCriterion",
]


class BertPretrainedModel(PretrainedModel):
    config_class = BertConfig
    base_model_prefix = "bert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class BertModel(BertPretrainedModel):
    """Embeddings + encoder + pooler (reference BertModel:270-427)."""

    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, task_type_ids=None,
                output_hidden_states=False):
        x = self.embeddings(input_ids, token_type_ids, position_ids)
        out = self.encoder(x, attention_mask, output_hidden_states)
        if output_hidden_states:
            sequence_output, all_hidden = out
        else:
            sequence_output, all_hidden = out, None
        pooled = self.pooler(sequence_output)
        if output_hidden_states:
            return sequence_output, pooled, all_hidden
        return sequence_output, pooled


class BertForSequenceClassification(BertPretrainedModel):
    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        _, pooled = self.bert(input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            if self.config.num_labels == 1:
                loss = F.mse_loss(logits.squeeze(-1), labels.float())
            else:
                loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class BertForTokenClassification(BertPretrainedModel):
    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.bert(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(sequence_output))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class BertForQuestionAnswering(BertPretrainedModel):
    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, start_positions=None, end_positions=None):
        sequence_output, _ = self.bert(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(sequence_output)
        start_logits, end_logits = logits.unbind(-1)
        if start_positions is not None and end_positions is not None:
            S = start_logits.shape[1]
            start_positions = start_positions.clamp(0, S - 1)
            end_positions = end_positions.clamp(0, S - 1)
            loss = 0.5 * (F.cross_entropy(start_logits, start_positions)
                          + F.cross_entropy(end_logits, end_positions))
            return loss, start_logits, end_logits
        return start_logits, end_logits


class BertForMultipleChoice(BertPretrainedModel):
    """input_ids: [B, num_choices, S] -> one logit per choice."""

    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, 1)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        B, C, S = input_ids.shape
        flat = lambda t: None if t is None else t.reshape(B * C, S)
        _, pooled = self.bert(flat(input_ids), flat(token_type_ids),
                              flat(position_ids), flat(attention_mask))
        logits = self.classifier(self.dropout(pooled)).view(B, C)
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class BertForMaskedLM(BertPretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.cls = LMPredictionHead(
            config, self.bert.embeddings.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.bert(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.cls(sequence_output)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class BertPretrainingHeads(nn.Module):
    """MLM head + next-sentence-prediction head (reference :892-941)."""

    def __init__(self, config, embedding_weights=None):
        super().__init__()
        self.predictions = LMPredictionHead(config, embedding_weights)
        self.seq_relationship = nn.Linear(config.hidden_size, 2)

    def forward(self, sequence_output, pooled_output):
        return self.predictions(sequence_output), self.seq_relationship(pooled_output)


class BertForPretraining(BertPretrainedModel):
    def __init__(self, config: BertConfig):
        super().__init__(config)
        self.bert = BertModel(config)
        self.cls = BertPretrainingHeads(
            config, self.bert.embeddings.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None, next_sentence_label=None):
        sequence_output, pooled = self.bert(
            input_ids, token_type_ids, position_ids, attention_mask)
        prediction_logits, seq_relationship_logits = self.cls(sequence_output, pooled)
        if labels is not None and next_sentence_label is not None:
            loss = BertPretrainingCriterion(self.config.vocab_size)(
                prediction_logits, seq_relationship_logits, labels,
                next_sentence_label)
            return loss, prediction_logits, seq_relationship_logits
        return prediction_logits, seq_relationship_logits


class BertPretrainingCriterion(nn.Module):
    """MLM CE (ignore -100) + NSP CE (reference :1084-1136)."""

    def __init__(self, vocab_size: int):
        super().__init__()
        self.vocab_size = vocab_size

    def forward(self, prediction_logits, seq_relationship_logits,
                masked_lm_labels, next_sentence_label):
        mlm = F.cross_entropy(
            prediction_logits.view(-1, self.vocab_size),
            masked_lm_labels.view(-1), ignore_index=-100)
        nsp = F.cross_entropy(seq_relationship_logits, next_sentence_label.view(-1))
        return mlm + nsp
