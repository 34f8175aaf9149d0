"""ChatGLM2 config (reference: paddlenlp/transformers/chatglm_v2/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["ChatGLMv2Config"]


class ChatGLMv2Config(PretrainedConfig):
    model_type = "chatglm_v2"

    def __init__(self, vocab_size=65024, hidden_size=4096,
                 num_hidden_layers=28, num_attention_heads=32,
                 multi_query_group_num=2, ffn_hidden_size=13696,
                 kv_channels=128, layernorm_epsilon=1e-5,
                 add_qkv_bias=True, initializer_range=0.02,
                 rope_theta=10000.0, max_position_embeddings=32768,
                 pad_token_id=0, bos_token_id=1, eos_token_id=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.multi_query_group_num = multi_query_group_num  # = kv heads
        self.ffn_hidden_size = ffn_hidden_size
        self.kv_channels = kv_channels
        self.layernorm_epsilon = layernorm_epsilon
        self.add_qkv_bias = add_qkv_bias
        self.initializer_range = initializer_range
        self.rope_theta = rope_theta
        self.max_position_embeddings = max_position_embeddings
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.rms_norm_eps = layernorm_epsilon

    @property
    def num_key_value_heads(self):
        return self.multi_query_group_num

    @property
    def head_dim(self):
        return self.kv_channels
