"""Mamba config (reference: paddlenlp/transformers/mamba/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["MambaConfig"]


class MambaConfig(PretrainedConfig):
    model_type = "mamba"

    def __init__(self, vocab_size=50280, hidden_size=768,
                 num_hidden_layers=24, state_size=16, conv_kernel=4,
                 expand=2, time_step_rank="auto", layer_norm_epsilon=1e-5,
                 initializer_range=0.1, use_bias=False, use_conv_bias=True,
                 pad_token_id=0, bos_token_id=0, eos_token_id=0,
                 tie_word_embeddings=True, **kwargs):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.state_size = state_size
        self.conv_kernel = conv_kernel
        self.expand = expand
        self.intermediate_size = expand * hidden_size
        self.time_step_rank = (
            max(1, hidden_size // 16) if time_step_rank == "auto" else time_step_rank)
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.use_bias = use_bias
        self.use_conv_bias = use_conv_bias
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.rms_norm_eps = layer_norm_epsilon
