"""Artist (reference: paddlenlp/transformers/artist/modeling.py).

Text-to-image GPT: the reference subclasses GPTModel/GPTLMHeadModel
(:36-42) with an enlarged vocabulary covering text tokens followed by
VQ image tokens, trained left-to-right over [text; image] sequences.
The classes here are the same thin subclassing over the framework GPT.
"""
from __future__ import annotations

from ..gpt.modeling import GPTConfig, GPTForCausalLM, GPTModel

__all__ = ["ArtistConfig", "ArtistModel", "ArtistForConditionalGeneration"]


class ArtistConfig(GPTConfig):
    model_type = "artist"

    def __init__(self, image_vocab_size=16384, image_length=256, **kwargs):
        super().__init__(**kwargs)
        self.image_vocab_size = image_vocab_size
        self.image_length = image_length


class ArtistModel(GPTModel):
    config_class = ArtistConfig
    base_model_prefix = "artist"


class ArtistForConditionalGeneration(GPTForCausalLM):
    config_class = ArtistConfig
    base_model_prefix = "artist"
