"""QWen (v1) config (reference: paddlenlp/transformers/qwen/configuration.py)."""
from ..configuration_utils import PretrainedConfig


class QWenConfig(PretrainedConfig):
    model_type = "qwen"

    def __init__(self, vocab_size=151936, hidden_size=4096,
                 num_hidden_layers=32, num_attention_heads=32,
                 intermediate_size=22016, layer_norm_epsilon=1e-6,
                 max_position_embeddings=8192, seq_length=2048,
                 rotary_emb_base=10000, use_logn_attn=False,
                 no_bias=True, initializer_range=0.02,
                 pad_token_id=None, eos_token_id=151643, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.layer_norm_epsilon = layer_norm_epsilon
        self.max_position_embeddings = max_position_embeddings
        self.seq_length = seq_length
        self.rotary_emb_base = rotary_emb_base
        self.use_logn_attn = use_logn_attn
        self.no_bias = no_bias
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
