"""BLOOM config (reference: paddlenlp/transformers/bloom/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["BloomConfig"]


class BloomConfig(PretrainedConfig):
    model_type = "bloom"

    attribute_map = {
        "n_embed": "hidden_size",
        "n_layer": "num_hidden_layers",
        "n_head": "num_attention_heads",
    }

    def __init__(self, vocab_size=250880, hidden_size=1024,
                 num_hidden_layers=24, num_attention_heads=16,
                 layer_norm_epsilon=1e-5, initializer_range=0.02,
                 apply_residual_connection_post_layernorm=False,
                 hidden_dropout=0.0, attention_dropout=0.0,
                 pad_token_id=3, bos_token_id=1, eos_token_id=2,
                 tie_word_embeddings=True, **kwargs):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.apply_residual_connection_post_layernorm = \
            apply_residual_connection_post_layernorm
        self.hidden_dropout = hidden_dropout
        self.attention_dropout = attention_dropout
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    @property
    def intermediate_size(self):
        return 4 * self.hidden_size
