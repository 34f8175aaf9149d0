"""Jamba config (reference: paddlenlp/transformers/jamba/configuration.py)."""
from ..configuration_utils import PretrainedConfig

__all__ = ["JambaConfig"]


class JambaConfig(PretrainedConfig):
    model_type = "jamba"

    def __init__(self, vocab_size=65536, hidden_size=4096,
                 intermediate_size=14336, num_hidden_layers=32,
                 num_attention_heads=32, num_key_value_heads=8,
                 rms_norm_eps=1e-6, initializer_range=0.02,
                 max_position_embeddings=262144,
                 # layer layout
                 attn_layer_period=8, attn_layer_offset=4,
                 expert_layer_period=2, expert_layer_offset=1,
                 num_experts=16, num_experts_per_tok=2,
                 router_aux_loss_coef=0.001,
                 # mamba mixer
                 mamba_d_state=16, mamba_d_conv=4, mamba_expand=2,
                 mamba_dt_rank="auto",
                 pad_token_id=0, bos_token_id=1, eos_token_id=2,
                 tie_word_embeddings=False, **kwargs):
        kwargs.setdefault("tie_word_embeddings", tie_word_embeddings)
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.num_key_value_heads = num_key_value_heads
        self.rms_norm_eps = rms_norm_eps
        self.initializer_range = initializer_range
        self.max_position_embeddings = max_position_embeddings
        self.attn_layer_period = attn_layer_period
        self.attn_layer_offset = attn_layer_offset
        self.expert_layer_period = expert_layer_period
        self.expert_layer_offset = expert_layer_offset
        self.num_experts = num_experts
        self.num_experts_per_tok = num_experts_per_tok
        self.router_aux_loss_coef = router_aux_loss_coef
        self.mamba_d_state = mamba_d_state
        self.mamba_d_conv = mamba_d_conv
        self.mamba_expand = mamba_expand
        self.mamba_dt_rank = (max(1, hidden_size // 16)
                              if mamba_dt_rank == "auto" else mamba_dt_rank)
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        # llama-attention knobs (jamba attention has NO rope: mamba carries
        # position); expose theta anyway for the shared attention class
        self.rope_theta = 10000.0
        self.rope_scaling_type = None
        self.rope_scaling_factor = 1.0
        self.fuse_attention_qkv = True

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads

    # adapter views for the reused Mamba mixer
    @property
    def state_size(self):
        return self.mamba_d_state

    @property
    def conv_kernel(self):
        return self.mamba_d_conv

    @property
    def time_step_rank(self):
        return self.mamba_dt_rank

    @property
    def use_bias(self):
        return False

    @property
    def use_conv_bias(self):
        return True
