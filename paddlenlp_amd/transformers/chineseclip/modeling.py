"""Chinese-CLIP (reference: paddlenlp/transformers/chineseclip/modeling.py).

CLIP with an asymmetric text tower: the vision side is the standard ViT
patch encoder, the TEXT side is a Chinese BERT (post-LN, bidirectional,
[CLS]-pooled — not CLIP's causal text transformer).  Projections and
temperature-scaled cosine logits as in CLIP.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..clip.modeling import CLIPVisionTransformer
from ..clip.configuration import CLIPVisionConfig
from ..encoder import EncoderLayer, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ChineseCLIPConfig", "ChineseCLIPModel"]


class ChineseCLIPTextConfig(PretrainedConfig):
    model_type = "chineseclip_text"

    def __init__(self, vocab_size=21128, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ChineseCLIPConfig(PretrainedConfig):
    model_type = "chineseclip"

    def __init__(self, text_config=None, vision_config=None,
                 projection_dim=512, logit_scale_init_value=2.6592,
                 **kwargs):
        super().__init__(**kwargs)
        self.text_config = ChineseCLIPTextConfig(**(text_config or {}))
        self.vision_config = CLIPVisionConfig(**(vision_config or {}))
        self.projection_dim = projection_dim
        self.logit_scale_init_value = logit_scale_init_value
        self.initializer_range = self.text_config.initializer_range


class ChineseCLIPPretrainedModel(PretrainedModel):
    config_class = ChineseCLIPConfig
    base_model_prefix = "chineseclip"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ChineseCLIPModel(ChineseCLIPPretrainedModel):
    def __init__(self, config: ChineseCLIPConfig):
        super().__init__(config)
        tc, vc = config.text_config, config.vision_config
        # BERT text tower
        self.text_embeddings = nn.Embedding(tc.vocab_size, tc.hidden_size,
                                            padding_idx=tc.pad_token_id)
        self.text_positions = nn.Embedding(tc.max_position_embeddings,
                                           tc.hidden_size)
        self.text_norm = nn.LayerNorm(tc.hidden_size, eps=tc.layer_norm_eps)
        self.text_layers = nn.ModuleList(
            [EncoderLayer(tc) for _ in range(tc.num_hidden_layers)])
        self.vision_model = CLIPVisionTransformer(vc)
        self.text_projection = nn.Linear(tc.hidden_size,
                                         config.projection_dim, bias=False)
        self.visual_projection = nn.Linear(vc.hidden_size,
                                           config.projection_dim, bias=False)
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
        return self.text_projection(x[:, 0])   # [CLS] pooling

    def get_image_features(self, pixel_values):
        _, pooled = self.vision_model(pixel_values)
        return self.visual_projection(pooled)

    def forward(self, input_ids, pixel_values, attention_mask=None):
        t = F.normalize(self.get_text_features(input_ids, attention_mask),
                        dim=-1)
        i = F.normalize(self.get_image_features(pixel_values), dim=-1)
        scale = self.logit_scale.exp()
        logits_per_text = scale * t @ i.t()
        return logits_per_text, logits_per_text.t()
