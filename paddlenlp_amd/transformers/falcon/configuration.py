"""Falcon (RW) config (reference: paddlenlp/transformers/rw/ falcon family)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["FalconConfig"]


class FalconConfig(PretrainedConfig):
    model_type = "falcon"

    def __init__(self, vocab_size=65024, hidden_size=4544,
                 num_hidden_layers=32, num_attention_heads=71,
                 num_key_value_heads=1, intermediate_size=None,
                 layer_norm_epsilon=1e-5, initializer_range=0.02,
                 rope_theta=10000.0, max_position_embeddings=2048,
                 parallel_attn=True, bias=False, pad_token_id=None,
                 bos_token_id=11, eos_token_id=11,
                 tie_word_embeddings=True, **kwargs):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.num_key_value_heads = num_key_value_heads
        self.intermediate_size = intermediate_size or 4 * hidden_size
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.rope_theta = rope_theta
        self.max_position_embeddings = max_position_embeddings
        self.parallel_attn = parallel_attn
        self.bias = bias
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        # llama attention path knobs
        self.rope_scaling_type = None
        self.rope_scaling_factor = 1.0
        self.fuse_attention_qkv = True
        self.rms_norm_eps = layer_norm_epsilon  # unused; attention only

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
