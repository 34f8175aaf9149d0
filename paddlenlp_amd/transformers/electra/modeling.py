"""ELECTRA model family (reference: paddlenlp/transformers/electra/modeling.py).

Encoder with a separate (smaller) embedding size projected up into the
hidden size, no pooler; heads: sequence/token classification, the
replaced-token-detection discriminator and the small-generator MLM head
used by ElectraForTotalPretraining.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..encoder import (
    ACT2FN,
    EncoderEmbeddings,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel
from .configuration import ElectraConfig

__all__ = [
    "ElectraModel",
    "ElectraPretrainedModel",
    "ElectraForSequenceClassification",
    "ElectraForTokenClassification",
    "ElectraDiscriminatorPredictions",
    "ElectraGeneratorPredictions",
    "ElectraDiscriminator",
    "ElectraGenerator",
    "ElectraForTotalPretraining",
]


class ElectraPretrainedModel(PretrainedModel):
    config_class = ElectraConfig
    base_model_prefix = "electra"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ElectraModel(ElectraPretrainedModel):
    """Embeddings (embedding_size) -> optional projection -> encoder; no pooler."""

    def __init__(self, config: ElectraConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.embeddings_project = (
            nn.Linear(config.embedding_size, config.hidden_size)
            if config.embedding_size != config.hidden_size else None)
        self.encoder = TransformerEncoder(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None):
        x = self.embeddings(input_ids, token_type_ids, position_ids)
        if self.embeddings_project is not None:
            x = self.embeddings_project(x)
        return self.encoder(x, attention_mask)


class ElectraForSequenceClassification(ElectraPretrainedModel):
    """GELU-activated two-layer head over the first token (no pooler)."""

    def __init__(self, config: ElectraConfig):
        super().__init__(config)
        self.electra = ElectraModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.dropout = nn.Dropout(p)
        self.out_proj = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output = self.electra(
            input_ids, token_type_ids, position_ids, attention_mask)
        x = self.dropout(sequence_output[:, 0])
        x = F.gelu(self.dense(x))
        logits = self.out_proj(self.dropout(x))
        if labels is not None:
            loss = F.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class ElectraForTokenClassification(ElectraPretrainedModel):
    def __init__(self, config: ElectraConfig):
        super().__init__(config)
        self.electra = ElectraModel(config)
        p = (config.classifier_dropout if config.classifier_dropout is not None
             else config.hidden_dropout_prob)
        self.dropout = nn.Dropout(p)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output = self.electra(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.classifier(self.dropout(sequence_output))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class ElectraDiscriminatorPredictions(nn.Module):
    """dense + act + dense(1): per-token replaced/original logit."""

    def __init__(self, config: ElectraConfig):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)
        self.dense_prediction = nn.Linear(config.hidden_size, 1)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, sequence_output):
        return self.dense_prediction(self.act(self.dense(sequence_output))).squeeze(-1)


class ElectraGeneratorPredictions(nn.Module):
    """dense to embedding_size + LN; decoder tied to word embeddings."""

    def __init__(self, config: ElectraConfig):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size, config.embedding_size)
        self.layer_norm = nn.LayerNorm(config.embedding_size, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, sequence_output):
        return self.layer_norm(self.act(self.dense(sequence_output)))


class ElectraDiscriminator(ElectraPretrainedModel):
    def __init__(self, config: ElectraConfig):
        super().__init__(config)
        self.electra = ElectraModel(config)
        self.discriminator_predictions = ElectraDiscriminatorPredictions(config)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output = self.electra(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.discriminator_predictions(sequence_output)
        if labels is not None:
            loss = F.binary_cross_entropy_with_logits(logits, labels.float())
            return loss, logits
        return logits


class ElectraGenerator(ElectraPretrainedModel):
    _tied_weights_keys = ["generator_lm_head.weight"]

    def __init__(self, config: ElectraConfig):
        super().__init__(config)
        self.electra = ElectraModel(config)
        self.generator_predictions = ElectraGeneratorPredictions(config)
        self.generator_lm_head = nn.Linear(
            config.embedding_size, config.vocab_size, bias=True)
        self.generator_lm_head.weight = self.electra.embeddings.word_embeddings.weight

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                attention_mask=None, labels=None):
        sequence_output = self.electra(
            input_ids, token_type_ids, position_ids, attention_mask)
        logits = self.generator_lm_head(self.generator_predictions(sequence_output))
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class ElectraForTotalPretraining(ElectraPretrainedModel):
    """Generator MLM + discriminator RTD joint objective
    (reference ElectraForTotalPretraining): the generator fills the masked
    positions, the discriminator labels each token replaced/original, and the
    losses combine as gen_weight * mlm + disc_weight * rtd."""

    def __init__(self, generator: ElectraGenerator, discriminator: ElectraDiscriminator):
        super().__init__(discriminator.config)
        self.generator = generator
        self.discriminator = discriminator

    def forward(self, input_ids, labels, token_type_ids=None,
                attention_mask=None):
        gen_loss, gen_logits = self.generator(
            input_ids, token_type_ids=token_type_ids,
            attention_mask=attention_mask, labels=labels)
        with torch.no_grad():
            sampled = gen_logits.argmax(-1)
        masked = labels != -100
        disc_input = torch.where(masked, sampled, input_ids)
        # a token is "replaced" if it was masked and the sample differs
        disc_labels = (masked & (sampled != labels)).long()
        disc_loss, disc_logits = self.discriminator(
            disc_input, token_type_ids=token_type_ids,
            attention_mask=attention_mask, labels=disc_labels)
        cfg = self.discriminator.config
        loss = cfg.gen_weight * gen_loss + cfg.disc_weight * disc_loss
        return loss, gen_logits, disc_logits
