"""LayoutLM family (reference: paddlenlp/transformers/layoutlm/modeling.py).

Document-AI encoder: BERT tower whose embeddings add 2-D layout tables —
x/y coordinate embeddings for all four bbox corners plus width/height
(h = x2-x0, w = y2-y0 in the reference's ordering) — over the standard
word/position/token_type sum.  bbox coords are 0..max_2d_position-1.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..bert.configuration import BertConfig
from ..encoder import EncoderPooler, TransformerEncoder, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["LayoutLMConfig", "LayoutLMModel",
           "LayoutLMForSequenceClassification",
           "LayoutLMForTokenClassification"]


class LayoutLMConfig(BertConfig):
    model_type = "layoutlm"

    def __init__(self, max_2d_position_embeddings: int = 1024, **kwargs):
        super().__init__(**kwargs)
        self.max_2d_position_embeddings = max_2d_position_embeddings


class LayoutLMPretrainedModel(PretrainedModel):
    config_class = LayoutLMConfig
    base_model_prefix = "layoutlm"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class LayoutLMEmbeddings(nn.Module):
    def __init__(self, config: LayoutLMConfig):
        super().__init__()
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h,
                                            padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.x_position_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, h)
        self.y_position_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, h)
        self.h_position_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, h)
        self.w_position_embeddings = nn.Embedding(
            config.max_2d_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.layer_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, input_ids, bbox=None, token_type_ids=None):
        B, S = input_ids.shape
        if token_type_ids is None:
            token_type_ids = torch.zeros_like(input_ids)
        pos = torch.arange(S, device=input_ids.device)
        x = self.word_embeddings(input_ids) \
            + self.position_embeddings(pos) \
            + self.token_type_embeddings(token_type_ids)
        if bbox is not None:  # [B, S, 4] = (x0, y0, x1, y1)
            x = (x
                 + self.x_position_embeddings(bbox[:, :, 0])
                 + self.y_position_embeddings(bbox[:, :, 1])
                 + self.x_position_embeddings(bbox[:, :, 2])
                 + self.y_position_embeddings(bbox[:, :, 3])
                 + self.h_position_embeddings(bbox[:, :, 3] - bbox[:, :, 1])
                 + self.w_position_embeddings(bbox[:, :, 2] - bbox[:, :, 0]))
        return self.dropout(self.layer_norm(x))


class LayoutLMModel(LayoutLMPretrainedModel):
    def __init__(self, config: LayoutLMConfig):
        super().__init__(config)
        self.embeddings = LayoutLMEmbeddings(config)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, bbox=None, token_type_ids=None,
                attention_mask=None):
        x = self.embeddings(input_ids, bbox, token_type_ids)
        seq = self.encoder(x, attention_mask)
        return seq, self.pooler(seq)


class LayoutLMForSequenceClassification(LayoutLMPretrainedModel):
    def __init__(self, config: LayoutLMConfig):
        super().__init__(config)
        self.layoutlm = LayoutLMModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, bbox=None, token_type_ids=None,
                attention_mask=None, labels=None):
        _, pooled = self.layoutlm(input_ids, bbox, token_type_ids,
                                  attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class LayoutLMForTokenClassification(LayoutLMPretrainedModel):
    def __init__(self, config: LayoutLMConfig):
        super().__init__(config)
        self.layoutlm = LayoutLMModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, bbox=None, token_type_ids=None,
                attention_mask=None, labels=None):
        seq, _ = self.layoutlm(input_ids, bbox, token_type_ids,
                               attention_mask)
        logits = self.classifier(self.dropout(seq))
        if labels is not None:
            loss = F.cross_entropy(logits.view(-1, logits.shape[-1]),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return logits
