"""Mistral config (reference: paddlenlp/transformers/mistral/configuration.py)."""
from ..llama.configuration import LlamaConfig

__all__ = ["MistralConfig"]


class MistralConfig(LlamaConfig):
    model_type = "mistral"

    def __init__(self, sliding_window=None, **kwargs):
        kwargs.setdefault("rms_norm_eps", 1e-5)
        kwargs.setdefault("rope_theta", 10000.0)
        super().__init__(**kwargs)
        self.sliding_window = sliding_window
