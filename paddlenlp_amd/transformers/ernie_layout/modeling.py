"""ERNIE-Layout model family (reference: paddlenlp/transformers/
ernie_layout/modeling.py) — document understanding: token embeddings plus
2-D spatial position embeddings of each token's bounding box (x0, y0, x1,
y1 in a 0..max_2d grid, plus width/height), over the shared encoder core.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ..encoder import EncoderEmbeddings, EncoderPooler, TransformerEncoder, \
    init_encoder_weights
from ..model_utils import PretrainedModel
from .configuration import ErnieLayoutConfig

__all__ = ["ErnieLayoutModel", "ErnieLayoutForQuestionAnswering",
           "ErnieLayoutForTokenClassification"]


class SpatialEmbeddings(nn.Module):
    """x/y/height/width embeddings of the token bbox (reference
    _cal_spatial_position_embeddings :133)."""

    def __init__(self, config: ErnieLayoutConfig):
        super().__init__()
        n = config.max_2d_position_embeddings
        h = config.hidden_size
        self.x_position_embeddings = nn.Embedding(n, h)
        self.y_position_embeddings = nn.Embedding(n, h)
        self.h_position_embeddings = nn.Embedding(n, h)
        self.w_position_embeddings = nn.Embedding(n, h)
        self.max_2d = n

    def forward(self, bbox):  # [B, S, 4] = (x0, y0, x1, y1)
        bbox = bbox.clamp(0, self.max_2d - 1)
        left = self.x_position_embeddings(bbox[:, :, 0])
        upper = self.y_position_embeddings(bbox[:, :, 1])
        right = self.x_position_embeddings(bbox[:, :, 2])
        lower = self.y_position_embeddings(bbox[:, :, 3])
        width = self.w_position_embeddings(
            (bbox[:, :, 2] - bbox[:, :, 0]).clamp(0, self.max_2d - 1))
        height = self.h_position_embeddings(
            (bbox[:, :, 3] - bbox[:, :, 1]).clamp(0, self.max_2d - 1))
        return left + upper + right + lower + width + height


class ErnieLayoutPretrainedModel(PretrainedModel):
    config_class = ErnieLayoutConfig
    base_model_prefix = "ernie_layout"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieLayoutModel(ErnieLayoutPretrainedModel):
    def __init__(self, config: ErnieLayoutConfig):
        super().__init__(config)
        self.embeddings = EncoderEmbeddings(config)
        self.spatial = SpatialEmbeddings(config)
        self.encoder = TransformerEncoder(config)
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings.word_embeddings

    def forward(self, input_ids, bbox=None, token_type_ids=None,
                position_ids=None, attention_mask=None):
        x = self.embeddings(input_ids, token_type_ids, position_ids)
        if bbox is not None:
            x = x + self.spatial(bbox)
        sequence_output = self.encoder(x, attention_mask)
        return sequence_output, self.pooler(sequence_output)


class ErnieLayoutForQuestionAnswering(ErnieLayoutPretrainedModel):
    """Extractive doc QA: start/end span over the document tokens."""

    def __init__(self, config: ErnieLayoutConfig):
        super().__init__(config)
        self.ernie_layout = ErnieLayoutModel(config)
        self.qa_outputs = nn.Linear(config.hidden_size, 2)

    def forward(self, input_ids, bbox=None, token_type_ids=None,
                attention_mask=None, start_positions=None, end_positions=None):
        sequence_output, _ = self.ernie_layout(
            input_ids, bbox, token_type_ids, None, attention_mask)
        start_logits, end_logits = self.qa_outputs(sequence_output).unbind(-1)
        if start_positions is not None and end_positions is not None:
            S = start_logits.shape[1]
            loss = 0.5 * (
                F.cross_entropy(start_logits, start_positions.clamp(0, S - 1))
                + F.cross_entropy(end_logits, end_positions.clamp(0, S - 1)))
            return loss, start_logits, end_logits
        return start_logits, end_logits


class ErnieLayoutForTokenClassification(ErnieLayoutPretrainedModel):
    def __init__(self, config: ErnieLayoutConfig):
        super().__init__(config)
        self.ernie_layout = ErnieLayoutModel(config)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, bbox=None, token_type_ids=None,
                attention_mask=None, labels=None):
        sequence_output, _ = self.ernie_layout(
            input_ids, bbox, token_type_ids, None, attention_mask)
        logits = self.classifier(sequence_output)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.num_labels), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
