"""mT5 family (reference: paddlenlp/transformers/mt5/modeling.py).

T5 v1.1 architecture with multilingual defaults: gated-GELU feed-forward,
no tied LM head, 250k sentencepiece vocab.  Re-exports the T5 stack with
those defaults baked in.
"""
from __future__ import annotations

from ..t5.configuration import T5Config
from ..t5.modeling import T5EncoderModel, T5ForConditionalGeneration, T5Model

__all__ = ["MT5Config", "MT5Model", "MT5EncoderModel",
           "MT5ForConditionalGeneration"]


class MT5Config(T5Config):
    model_type = "mt5"

    def __init__(self, vocab_size=250112, d_model=512, d_kv=64, d_ff=1024,
                 num_layers=8, num_heads=6, feed_forward_proj="gated-gelu",
                 tie_word_embeddings=False, **kwargs):
        super().__init__(
            vocab_size=vocab_size, d_model=d_model, d_kv=d_kv, d_ff=d_ff,
            num_layers=num_layers, num_heads=num_heads,
            feed_forward_proj=feed_forward_proj,
            tie_word_embeddings=tie_word_embeddings, **kwargs)


class MT5Model(T5Model):
    config_class = MT5Config


class MT5EncoderModel(T5EncoderModel):
    config_class = MT5Config


class MT5ForConditionalGeneration(T5ForConditionalGeneration):
    config_class = MT5Config
