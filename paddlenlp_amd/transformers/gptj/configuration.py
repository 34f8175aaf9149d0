"""GPT-J config (reference: paddlenlp/transformers/gptj/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["GPTJConfig"]


class GPTJConfig(PretrainedConfig):
    model_type = "gptj"

    attribute_map = {
        "n_embd": "hidden_size",
        "n_layer": "num_hidden_layers",
        "n_head": "num_attention_heads",
    }

    def __init__(self, vocab_size=50400, hidden_size=4096,
                 num_hidden_layers=28, num_attention_heads=16,
                 rotary_dim=64, intermediate_size=None,
                 layer_norm_epsilon=1e-5, initializer_range=0.02,
                 rope_theta=10000.0, max_position_embeddings=2048,
                 pad_token_id=None, bos_token_id=50256, eos_token_id=50256,
                 tie_word_embeddings=False, **kwargs):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.rotary_dim = rotary_dim
        self.intermediate_size = intermediate_size or 4 * hidden_size
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.rope_theta = rope_theta
        self.max_position_embeddings = max_position_embeddings
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
