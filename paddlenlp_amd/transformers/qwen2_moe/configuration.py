"""Qwen2-MoE config (reference: paddlenlp/transformers/qwen2_moe/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["Qwen2MoeConfig"]


class Qwen2MoeConfig(PretrainedConfig):
    model_type = "qwen2_moe"

    def __init__(
        self,
        vocab_size=151936,
        hidden_size=2048,
        intermediate_size=5632,
        num_hidden_layers=24,
        num_attention_heads=16,
        num_key_value_heads=16,
        max_position_embeddings=32768,
        initializer_range=0.02,
        rms_norm_eps=1e-6,
        rope_theta=1e6,
        attention_bias=True,
        num_experts=60,
        num_experts_per_tok=4,
        moe_intermediate_size=1408,
        shared_expert_intermediate_size=5632,
        norm_topk_prob=False,
        decoder_sparse_step=1,
        mlp_only_layers=None,
        router_aux_loss_coef=0.001,
        output_router_logits=False,
        expert_parallel_degree=1,
        pad_token_id=None,
        bos_token_id=151643,
        eos_token_id=151643,
        tie_word_embeddings=False,
        **kwargs,
    ):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
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
        self.attention_bias = attention_bias
        self.num_experts = num_experts
        self.num_experts_per_tok = num_experts_per_tok
        self.moe_intermediate_size = moe_intermediate_size
        self.shared_expert_intermediate_size = shared_expert_intermediate_size
        self.norm_topk_prob = norm_topk_prob
        self.decoder_sparse_step = decoder_sparse_step
        self.mlp_only_layers = mlp_only_layers or []
        self.router_aux_loss_coef = router_aux_loss_coef
        self.output_router_logits = output_router_logits
        self.expert_parallel_degree = expert_parallel_degree
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        # llama attention path knobs
        self.rope_scaling_type = kwargs.get("rope_scaling_type", None)
        self.rope_scaling_factor = kwargs.get("rope_scaling_factor", 1.0)
        self.fuse_attention_qkv = kwargs.get("fuse_attention_qkv", True)

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads
