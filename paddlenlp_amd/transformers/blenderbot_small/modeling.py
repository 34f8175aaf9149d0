"""BlenderbotSmall (reference: paddlenlp/transformers/blenderbot_small/).

Same dialogue seq2seq as blenderbot but POST-LN (normalize_before=False,
bart-style layers, no final stack layernorm) — 90M-scale checkpoints.
"""
from __future__ import annotations

from ..bart.modeling import BartDecoderLayer, BartEncoderLayer
from ..blenderbot.modeling import (
    BlenderbotConfig,
    BlenderbotForConditionalGeneration,
    BlenderbotModel,
    _BlenderbotDecoder,
    _BlenderbotEncoder,
)

__all__ = ["BlenderbotSmallConfig", "BlenderbotSmallModel",
           "BlenderbotSmallForConditionalGeneration"]


class BlenderbotSmallConfig(BlenderbotConfig):
    model_type = "blenderbot_small"

    def __init__(self, normalize_before=False, scale_embedding=True, **kwargs):
        kwargs["normalize_before"] = normalize_before
        kwargs["scale_embedding"] = scale_embedding
        super().__init__(**kwargs)


class _SmallEncoder(_BlenderbotEncoder):
    encoder_layer_cls = BartEncoderLayer


class _SmallDecoder(_BlenderbotDecoder):
    decoder_layer_cls = BartDecoderLayer


class BlenderbotSmallModel(BlenderbotModel):
    config_class = BlenderbotSmallConfig
    base_model_prefix = "blenderbot_small"
    encoder_cls = _SmallEncoder
    decoder_cls = _SmallDecoder


class BlenderbotSmallForConditionalGeneration(BlenderbotForConditionalGeneration):
    config_class = BlenderbotSmallConfig
    base_model_prefix = "blenderbot_small"
    model_cls = BlenderbotSmallModel
    base_attr = "blenderbot_small"
    _tied_weights_keys = ["blenderbot_small.encoder.embed_tokens.weight",
                          "blenderbot_small.decoder.embed_tokens.weight",
                          "lm_head.weight"]
