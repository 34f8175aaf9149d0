"""Qwen2 config (reference: paddlenlp/transformers/qwen2/configuration.py)."""
from ..llama.configuration import LlamaConfig

__all__ = ["Qwen2Config"]


class Qwen2Config(LlamaConfig):
    """Llama architecture + attention-projection bias (qkv_bias)."""

    model_type = "qwen2"

    def __init__(self, attention_bias=True, tie_word_embeddings=False, **kwargs):
        kwargs.setdefault("rms_norm_eps", 1e-6)
        kwargs.setdefault("rope_theta", 1e6)
        kwargs["tie_word_embeddings"] = tie_word_embeddings
        super().__init__(**kwargs)
        self.attention_bias = attention_bias
