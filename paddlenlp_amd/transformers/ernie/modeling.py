"""ERNIE model family (reference: paddlenlp/transformers/ernie/modeling.py).

BERT-architecture encoder with optional task-type embeddings, plus the UIE
(Universal Information Extraction) pointer head that powers the Taskflow
information-extraction pipelines.  Built on the shared encoder core.
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
from .configuration import ErnieConfig

__all__ = [
    "ErnieModel",
    "ErniePretrainedModel",
    "ErnieForSequenceClassification",
    "ErnieForTokenClassification",
    "ErnieForQuestionAnswering",
    "ErnieForMaskedLM",
    "ErnieForPretraining",
    "ErniePretrainingCriterion",
    "UIE",
    "UTC",
]


class ErniePretrainedModel(PretrainedModel):
    config_class = ErnieConfig
    base_model_prefix = "ernie"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieModel(ErniePretrainedModel):
    """reference ErnieModel:183-381 (task_type embeddings when use_task_id)."""

    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, task_type_ids=None):
        if task_type_ids is None and self.config.use_task_id:
            task_type_ids = torch.full_like(input_ids, self.config.task_id)
        x = self.embeddings(input_ids, token_type_ids, position_ids, task_type_ids)
        sequence_output = self.encoder(x, attention_mask)
        return sequence_output, self.pooler(sequence_output)


class _ClassifierBase(ErniePretrainedModel):
    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.ernie = ErnieModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)


class ErnieForSequenceClassification(_ClassifierBase):
    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        _, pooled = self.ernie(input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            if self.config.num_labels == 1:
                loss = F.mse_loss(logits.squeeze(-1), labels.float())
            else:
                loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class ErnieForTokenClassification(_ClassifierBase):
    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.ernie(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(sequence_output))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class ErnieForQuestionAnswering(ErniePretrainedModel):
    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.ernie = ErnieModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, start_positions=None, end_positions=None):
        sequence_output, _ = self.ernie(
            input_ids, token_type_ids, position_ids, attention_mask)
        start_logits, end_logits = self.classifier(sequence_output).unbind(-1)
        if start_positions is not None and end_positions is not None:
            S = start_logits.shape[1]
            loss = 0.5 * (
                F.cross_entropy(start_logits, start_positions.clamp(0, S - 1))
                + F.cross_entropy(end_logits, end_positions.clamp(0, S - 1)))
            return loss, start_logits, end_logits
        return start_logits, end_logits


class ErnieForMaskedLM(ErniePretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.ernie = ErnieModel(config)
        self.cls = LMPredictionHead(
            config, self.ernie.embeddings.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.ernie(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.cls(sequence_output)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class ErniePretrainingHeads(nn.Module):
    def __init__(self, config, embedding_weights=None):
        super().__init__()
        self.predictions = LMPredictionHead(config, embedding_weights)
        self.seq_relationship = nn.Linear(config.hidden_size, 2)

    def forward(self, sequence_output, pooled_output):
        return self.predictions(sequence_output), self.seq_relationship(pooled_output)


class ErnieForPretraining(ErniePretrainedModel):
    """MLM + sentence-order/NSP pretraining (reference :820-931)."""

    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.ernie = ErnieModel(config)
        self.cls = ErniePretrainingHeads(
            config, self.ernie.embeddings.word_embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None, next_sentence_label=None):
        sequence_output, pooled = self.ernie(
            input_ids, token_type_ids, position_ids, attention_mask)
        prediction_logits, seq_logits = self.cls(sequence_output, pooled)
        if labels is not None and next_sentence_label is not None:
            loss = ErniePretrainingCriterion(self.config.vocab_size)(
                prediction_logits, seq_logits, labels, next_sentence_label)
            return loss, prediction_logits, seq_logits
        return prediction_logits, seq_logits


class ErniePretrainingCriterion(nn.Module):
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


class UTC(ErniePretrainedModel):
    """Unified Tag Classification (reference :1286-1384): a query projection
    of the [CLS] state dotted against key projections of each option's
    [O-MASK] state gives one logit per candidate label — the backbone of
    zero-shot text classification."""

    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.ernie = ErnieModel(config)
        self.predict_size = 64
        self.linear_q = nn.Linear(config.hidden_size, self.predict_size)
        self.linear_k = nn.Linear(config.hidden_size, self.predict_size)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, omask_positions=None, cls_positions=None):
        """omask_positions [B, max_options] (0-padded); cls_positions [B]."""
        sequence_output, _ = self.ernie(
            input_ids, token_type_ids, position_ids, attention_mask)
        B = sequence_output.shape[0]
        batch_idx = torch.arange(B, device=sequence_output.device)
        if cls_positions is None:
            cls_positions = torch.zeros(B, dtype=torch.long,
                                        device=sequence_output.device)
        q = self.linear_q(sequence_output[batch_idx, cls_positions])  # [B, P]
        option_states = sequence_output[
            batch_idx[:, None], omask_positions]                       # [B, O, H]
        k = self.linear_k(option_states)                               # [B, O, P]
        logits = torch.einsum("bp,bop->bo", q, k) / self.predict_size ** 0.5
        return logits


class UIE(ErniePretrainedModel):
    """Universal Information Extraction pointer head (reference :1222-1284):
    sigmoid start/end probabilities over the sequence."""

    def __init__(self, config: ErnieConfig):
        super().__init__(config)
        self.ernie = ErnieModel(config)
        self.linear_start = nn.Linear(config.hidden_size, 1)
        self.linear_end = nn.Linear(config.hidden_size, 1)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None):
        sequence_output, _ = self.ernie(
            input_ids, token_type_ids, position_ids, attention_mask)
        start_prob = torch.sigmoid(self.linear_start(sequence_output).squeeze(-1))
        end_prob = torch.sigmoid(self.linear_end(sequence_output).squeeze(-1))
        return start_prob, end_prob
