"""OPT config (reference: paddlenlp/transformers/opt/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["OPTConfig"]


class OPTConfig(PretrainedConfig):
    model_type = "opt"

    def __init__(self, vocab_size=50272, hidden_size=768, num_hidden_layers=12,
                 num_attention_heads=12, intermediate_size=3072,
                 hidden_act="relu", max_position_embeddings=2048,
                 do_layer_norm_before=True, initializer_range=0.02,
                 layer_norm_epsilon=1e-5, pad_token_id=1, bos_token_id=2,
                 eos_token_id=2, tie_word_embeddings=True, **kwargs):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.max_position_embeddings = max_position_embeddings
        self.do_layer_norm_before = do_layer_norm_before
        self.initializer_range = initializer_range
        self.layer_norm_epsilon = layer_norm_epsilon
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
