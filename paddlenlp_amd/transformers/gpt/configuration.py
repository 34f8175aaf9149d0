"""GPT-2/3 config (reference: paddlenlp/transformers/gpt/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["GPTConfig"]


class GPTConfig(PretrainedConfig):
    model_type = "gpt"

    attribute_map = {
        "n_positions": "max_position_embeddings",
        "n_embd": "hidden_size",
        "n_layer": "num_hidden_layers",
        "n_head": "num_attention_heads",
        "n_inner": "intermediate_size",
    }

    def __init__(
        self,
        vocab_size=50304,
        hidden_size=768,
        intermediate_size=None,
        num_hidden_layers=12,
        num_attention_heads=12,
        hidden_act="gelu",
        max_position_embeddings=1024,
        initializer_range=0.02,
        layer_norm_epsilon=1e-5,
        attention_probs_dropout_prob=0.0,
        hidden_dropout_prob=0.0,
        pad_token_id=None,
        bos_token_id=50256,
        eos_token_id=50256,
        tie_word_embeddings=True,
        **kwargs,
    ):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size or 4 * hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.hidden_act = hidden_act
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.layer_norm_epsilon = layer_norm_epsilon
        self.attention_probs_dropout_prob = attention_probs_dropout_prob
        self.hidden_dropout_prob = hidden_dropout_prob
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
