"""DeepSeek-V2 config (reference: paddlenlp/transformers/deepseek_v2/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["DeepseekV2Config"]


class DeepseekV2Config(PretrainedConfig):
    model_type = "deepseek_v2"

    def __init__(
        self,
        vocab_size=102400,
        hidden_size=2048,
        intermediate_size=10944,
        moe_intermediate_size=1408,
        num_hidden_layers=27,
        num_attention_heads=16,
        max_position_embeddings=4096,
        initializer_range=0.02,
        rms_norm_eps=1e-6,
        rope_theta=10000.0,
        # MLA (multi-head latent attention)
        q_lora_rank=None,          # None: plain q projection (V2-Lite)
        kv_lora_rank=512,
        qk_nope_head_dim=128,
        qk_rope_head_dim=64,
        v_head_dim=128,
        # MoE
        n_routed_experts=None,     # None: dense model
        n_shared_experts=None,
        num_experts_per_tok=6,
        n_group=1,
        topk_group=1,
        routed_scaling_factor=1.0,
        first_k_dense_replace=1,
        moe_layer_freq=1,
        norm_topk_prob=False,
        aux_loss_alpha=0.001,
        expert_parallel_degree=1,
        pad_token_id=None,
        bos_token_id=100000,
        eos_token_id=100001,
        **kwargs,
    ):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.moe_intermediate_size = moe_intermediate_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.rms_norm_eps = rms_norm_eps
        self.rope_theta = rope_theta
        self.q_lora_rank = q_lora_rank
        self.kv_lora_rank = kv_lora_rank
        self.qk_nope_head_dim = qk_nope_head_dim
        self.qk_rope_head_dim = qk_rope_head_dim
        self.v_head_dim = v_head_dim
        self.n_routed_experts = n_routed_experts
        self.n_shared_experts = n_shared_experts
        self.num_experts_per_tok = num_experts_per_tok
        self.n_group = n_group
        self.topk_group = topk_group
        self.routed_scaling_factor = routed_scaling_factor
        self.first_k_dense_replace = first_k_dense_replace
        self.moe_layer_freq = moe_layer_freq
        self.norm_topk_prob = norm_topk_prob
        self.aux_loss_alpha = aux_loss_alpha
        self.expert_parallel_degree = expert_parallel_degree
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def qk_head_dim(self):
        return self.qk_nope_head_dim + self.qk_rope_head_dim

    @property
    def head_dim(self):
        return self.qk_head_dim
