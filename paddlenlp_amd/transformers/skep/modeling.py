"""SKEP sentiment family (reference: paddlenlp/transformers/skep/modeling.py).

Sentiment-Knowledge-Enhanced Pretraining encoder: BERT architecture with
type_vocab_size=4, plus the reference's three heads — sequence
classification, token classification, and the BiGRU+linear-chain-CRF tagger
(SkepCrfForTokenClassification) used for opinion extraction, built on
paddlenlp_amd.layers.crf.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ...layers import LinearChainCrf, LinearChainCrfLoss, ViterbiDecoder
from ..bert.configuration import BertConfig
from ..encoder import (
    EncoderEmbeddings,
    EncoderPooler,
    TransformerEncoder,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["SkepConfig", "SkepModel", "SkepForSequenceClassification",
           "SkepForTokenClassification", "SkepCrfForTokenClassification"]


class SkepConfig(BertConfig):
    model_type = "skep"

    def __init__(self, type_vocab_size: int = 4, **kwargs):
        kwargs["type_vocab_size"] = type_vocab_size
        super().__init__(**kwargs)


class SkepPretrainedModel(PretrainedModel):
    config_class = SkepConfig
    base_model_prefix = "skep"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class SkepModel(SkepPretrainedModel):
    def __init__(self, config: SkepConfig):
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


class SkepForSequenceClassification(SkepPretrainedModel):
    def __init__(self, config: SkepConfig):
        super().__init__(config)
        self.skep = SkepModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.skep(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return nn.functional.cross_entropy(logits, labels.view(-1)), logits
        return logits


class SkepForTokenClassification(SkepPretrainedModel):
    def __init__(self, config: SkepConfig):
        super().__init__(config)
        self.skep = SkepModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.skep(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(seq))
        if labels is not None:
            loss = nn.functional.cross_entropy(
                logits.view(-1, logits.shape[-1]), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits


class SkepCrfForTokenClassification(SkepPretrainedModel):
    """BiGRU + CRF tagger (reference skep/modeling.py:634; crf_lr=0.2,
    no start/stop tags)."""

    def __init__(self, config: SkepConfig):
        super().__init__(config)
        self.skep = SkepModel(config)
        gru_hidden = 128
        self.gru = nn.GRU(config.hidden_size, gru_hidden, num_layers=2,
                          batch_first=True, bidirectional=True)
        self.fc = nn.Linear(2 * gru_hidden, config.num_labels)
        self.crf = LinearChainCrf(config.num_labels, crf_lr=0.2,
                                  with_start_stop_tag=False)
        self.crf_loss = LinearChainCrfLoss(self.crf)
        self.viterbi_decoder = ViterbiDecoder(self.crf.transitions, False)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                seq_lens=None, labels=None):
        seq, _ = self.skep(input_ids, token_type_ids, attention_mask)
        if seq_lens is None:
            seq_lens = torch.full((input_ids.shape[0],), input_ids.shape[1],
                                  dtype=torch.long, device=input_ids.device)
        feats, _ = self.gru(seq)
        emissions = self.fc(feats)
        if labels is not None:
            return self.crf_loss(emissions, seq_lens, labels)
        _, paths = self.viterbi_decoder(emissions, seq_lens)
        return paths
