"""Gemma config (reference: paddlenlp/transformers/gemma/configuration.py)."""
from ..llama.configuration import LlamaConfig

__all__ = ["GemmaConfig"]


class GemmaConfig(LlamaConfig):
    model_type = "gemma"

    def __init__(self, vocab_size=256000, hidden_size=2048,
                 intermediate_size=16384, num_hidden_layers=18,
                 num_attention_heads=8, num_key_value_heads=1,
                 head_dim=256, hidden_act="gelu_tanh", rope_theta=10000.0,
                 max_position_embeddings=8192, rms_norm_eps=1e-6,
                 tie_word_embeddings=True, bos_token_id=2, eos_token_id=1,
                 pad_token_id=0, **kwargs):
        # round-trip: to_dict() serializes the private backing field
        head_dim = kwargs.pop("_head_dim", head_dim)
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(
            vocab_size=vocab_size, hidden_size=hidden_size,
            intermediate_size=intermediate_size,
            num_hidden_layers=num_hidden_layers,
            num_attention_heads=num_attention_heads,
            num_key_value_heads=num_key_value_heads,
            rope_theta=rope_theta,
            max_position_embeddings=max_position_embeddings,
            rms_norm_eps=rms_norm_eps, bos_token_id=bos_token_id,
            eos_token_id=eos_token_id, pad_token_id=pad_token_id, **kwargs)
        self._head_dim = head_dim
        self.hidden_act = hidden_act

    @property
    def head_dim(self):
        return self._head_dim
