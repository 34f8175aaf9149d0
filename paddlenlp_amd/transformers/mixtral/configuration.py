"""Mixtral config (reference: paddlenlp/transformers/mixtral/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["MixtralConfig"]


class MixtralConfig(PretrainedConfig):
    model_type = "mixtral"

    def __init__(
        self,
        vocab_size=32000,
        hidden_size=4096,
        intermediate_size=14336,
        num_hidden_layers=32,
        num_attention_heads=32,
        num_key_value_heads=8,
        max_position_embeddings=32768,
        initializer_range=0.02,
        rms_norm_eps=1e-5,
        rope_theta=1e6,
        num_local_experts=8,
        num_experts_per_tok=2,
        router_aux_loss_coef=0.02,
        output_router_logits=False,
        expert_parallel_degree=1,
        pad_token_id=None,
        bos_token_id=1,
        eos_token_id=2,
        **kwargs,
    ):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.num_key_value_heads = num_key_value_heads
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.rms_norm_eps = rms_norm_eps
        self.rope_theta = rope_theta
        self.num_local_experts = num_local_experts
        self.num_experts_per_tok = num_experts_per_tok
        self.router_aux_loss_coef = router_aux_loss_coef
        self.output_router_logits = output_router_logits
        self.expert_parallel_degree = expert_parallel_degree
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        # reuse the llama attention path
        self.rope_scaling_type = kwargs.get("rope_scaling_type", None)
        self.rope_scaling_factor = kwargs.get("rope_scaling_factor", 1.0)
        self.fuse_attention_qkv = kwargs.get("fuse_attention_qkv", True)

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
