"""ERNIE-Code (reference: paddlenlp/transformers/ernie_code/modeling.py).

Multilingual code/text seq2seq.  The reference file duplicates the full
T5/mT5 stack under the ErnieCode name (gated-act FF, relative-bucket
attention bias, T5 LayerNorm — its classes :80-560 are line-level
copies of mt5); here the model IS the shared T5 implementation with
ERNIE-Code's config defaults (gated-gelu, tie_word_embeddings=False).
"""
from __future__ import annotations

from ..mt5.modeling import (
    MT5Config,
    MT5EncoderModel,
    MT5ForConditionalGeneration,
    MT5Model,
)

__all__ = ["ErnieCodeConfig", "ErnieCodeModel",
           "ErnieCodeForConditionalGeneration", "ErnieCodeEncoderModel"]


class ErnieCodeConfig(MT5Config):
    model_type = "ernie_code"


class ErnieCodeModel(MT5Model):
    config_class = ErnieCodeConfig
    base_model_prefix = "ernie_code"


class ErnieCodeEncoderModel(MT5EncoderModel):
    config_class = ErnieCodeConfig
    base_model_prefix = "ernie_code"


class ErnieCodeForConditionalGeneration(MT5ForConditionalGeneration):
    config_class = ErnieCodeConfig
    base_model_prefix = "ernie_code"
