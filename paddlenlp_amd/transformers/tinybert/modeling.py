"""TinyBERT family (reference: paddlenlp/transformers/tinybert/modeling.py).

BERT-architecture student for two-stage distillation: the base model carries
``fit_dense`` / per-layer ``fit_denses`` linear projections that map the
student's hidden states (hidden_size) into the teacher's width (fit_size)
for the intermediate-layer distillation loss.  Heads mirror the reference:
pretraining (hidden-state output for distill), sequence classification and
QA.
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

__all__ = ["TinyBertConfig", "TinyBertModel", "TinyBertForPretraining",
           "TinyBertForSequenceClassification", "TinyBertForQuestionAnswering"]


class TinyBertConfig(BertConfig):
    model_type = "tinybert"

    def __init__(self, fit_size: int = 768, **kwargs):
        super().__init__(**kwargs)
        self.fit_size = fit_size


class TinyBertPretrainedModel(PretrainedModel):
    config_class = TinyBertConfig
    base_model_prefix = "tinybert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class TinyBertModel(TinyBertPretrainedModel):
    def __init__(self, config: TinyBertConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)
        # student->teacher width projections for hidden-state distillation
        self.fit_denses = nn.ModuleList([
            nn.Linear(config.hidden_size, config.fit_size)
            for _ in range(config.num_hidden_layers + 1)])
        self.fit_dense = nn.Linear(config.hidden_size, config.fit_size)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                output_hidden_states=False):
        x = self.embeddings(input_ids, token_type_ids)
        out = self.encoder(x, attention_mask, output_hidden_states)
        if output_hidden_states:
            sequence_output, all_hidden = out
        else:
            sequence_output, all_hidden = out, None
        pooled = self.pooler(sequence_output)
        if output_hidden_states:
            return sequence_output, pooled, all_hidden
        return sequence_output, pooled


class TinyBertForPretraining(TinyBertPretrainedModel):
    """Distillation-stage model: returns teacher-width hidden states."""

    def __init__(self, config: TinyBertConfig):
        super().__init__(config)
        self.tinybert = TinyBertModel(config)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, _, all_hidden = self.tinybert(
            input_ids, token_type_ids, attention_mask,
            output_hidden_states=True)
        return [self.tinybert.fit_denses[i](h) for i, h in enumerate(all_hidden)]


class TinyBertForSequenceClassification(TinyBertPretrainedModel):
    def __init__(self, config: TinyBertConfig):
        super().__init__(config)
        self.tinybert = TinyBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.tinybert(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            loss = nn.functional.cross_entropy(logits, labels.view(-1))
            return loss, logits
        return logits


class TinyBertForQuestionAnswering(TinyBertPretrainedModel):
    def __init__(self, config: TinyBertConfig):
        super().__init__(config)
        self.tinybert = TinyBertModel(config)
        self.classifier = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        seq, _ = self.tinybert(input_ids, token_type_ids, attention_mask)
        start, end = self.classifier(seq).chunk(2, dim=-1)
        return start.squeeze(-1), end.squeeze(-1)
