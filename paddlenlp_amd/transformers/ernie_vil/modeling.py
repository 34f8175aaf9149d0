"""ERNIE-ViL 2.0 (reference: paddlenlp/transformers/ernie_vil/modeling.py).

Dual-tower image-text contrastive model: an ERNIE text encoder and a
ViT image encoder, [CLS]/class-token pooled WITHOUT projection heads
(ERNIE-ViL 2.0 matches tower widths and compares pooled features
directly), temperature-scaled cosine logits.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..clip.configuration import CLIPVisionConfig
from ..clip.modeling import CLIPVisionTransformer
from ..configuration_utils import PretrainedConfig
from ..encoder import EncoderLayer, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ErnieViLConfig", "ErnieViLModel"]


class ErnieViLTextConfig(PretrainedConfig):
    model_type = "ernie_vil_text"

    def __init__(self, vocab_size=40000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 max_position_embeddings=2048, type_vocab_size=4,
                 layer_norm_eps=1e-12, initializer_range=0.02,
                 pad_token_id=0, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = 0.0
        self.attention_probs_dropout_prob = 0.0
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ErnieViLConfig(PretrainedConfig):
    model_type = "ernie_vil"

    def __init__(self, text_config=None, vision_config=None,
                 logit_scale_init_value=2.6592, **kwargs):
        super().__init__(**kwargs)
        self.text_config = ErnieViLTextConfig(**(text_config or {}))
        self.vision_config = CLIPVisionConfig(**(vision_config or {}))
        self.logit_scale_init_value = logit_scale_init_value
        self.initializer_range = self.text_config.initializer_range


class ErnieViLModel(PretrainedModel):
    config_class = ErnieViLConfig
    base_model_prefix = "ernie_vil"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)

    def __init__(self, config: ErnieViLConfig):
        super().__init__(config)
        tc = config.text_config
        self.text_embeddings = nn.Embedding(tc.vocab_size, tc.hidden_size,
                                            padding_idx=tc.pad_token_id)
        self.text_positions = nn.Embedding(tc.max_position_embeddings,
                                           tc.hidden_size)
        self.text_norm = nn.LayerNorm(tc.hidden_size, eps=tc.layer_norm_eps)
        self.text_layers = nn.ModuleList(
            [EncoderLayer(tc) for _ in range(tc.num_hidden_layers)])
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        assert tc.hidden_size == config.vision_config.hidden_size, \
            "ERNIE-ViL 2.0 compares tower features directly: widths must match"
        self.logit_scale = nn.Parameter(
            torch.tensor(config.logit_scale_init_value))
        self.init_weights()

    def get_input_embeddings(self):
        return self.text_embeddings

    def get_text_features(self, input_ids, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.text_norm(self.text_embeddings(input_ids) +
                           self.text_positions(pos))
        for layer in self.text_layers:
            x = layer(x, attention_mask)
        return x[:, 0]

    def get_image_features(self, pixel_values):
        _, pooled = self.vision_model(pixel_values)
        return pooled

    def forward(self, input_ids, pixel_values, attention_mask=None):
        t = F.normalize(self.get_text_features(input_ids, attention_mask),
                        dim=-1)
        i = F.normalize(self.get_image_features(pixel_values), dim=-1)
        scale = self.logit_scale.exp()
        logits_per_text = scale * t @ i.t()
        return logits_per_text, logits_per_text.t()
