"""Yuan-2.0 family (reference: paddlenlp/transformers/yuan/modeling.py).

Llama-shaped decoder whose attention inputs pass through LFA (Localized
Filtering-based Attention): a 2-tap causal conv pair over the sequence
(conv1: C -> C/2, conv2: C/2 -> C, both kernel (2,1)) with residual +
RMSNorm, applied to the Q/K path only (V reads the raw hidden states).
The decode cache therefore carries a third element: the last TWO raw hidden
states, so the conv window is exact at every step.  MLP is swiglu with the
gate/up roles swapped vs llama: down(gate(x) * silu(up(x))).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..model_utils import PretrainedModel

__all__ = ["YuanConfig", "YuanModel", "YuanForCausalLM"]


class YuanConfig(PretrainedConfig):
    model_type = "yuan"

    def __init__(self, vocab_size=135040, hidden_size=2048,
                 intermediate_size=8192, num_hidden_layers=24,
                 num_attention_heads=16, num_key_value_heads=None,
                 rms_norm_eps=1e-6, max_position_embeddings=8192,
                 rope_theta=10000.0, initializer_range=0.02,
                 pad_token_id=77185, bos_token_id=77185, eos_token_id=77185,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.intermediate_size = intermediate_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.num_key_value_heads = (num_key_value_heads
                                    if num_key_value_heads is not None
                                    else num_attention_heads)
        self.rms_norm_eps = rms_norm_eps
        self.max_position_embeddings = max_position_embeddings
        self.rope_theta = rope_theta
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class YuanRMSNorm(nn.Module):
    def __init__(self, hidden_size, eps=1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


class LocalizedFiltering(nn.Module):
    """2-tap causal conv chain + residual + RMSNorm (reference yuan:78).

    Weights are Conv2d with kernel (2, 1) to match the reference checkpoint
    layout; padding is done explicitly so train/prefill and cached decode
    share one code path.
    """

    def __init__(self, hidden_size):
        super().__init__()
        self.embed_dim = hidden_size
        self.conv1 = nn.Conv2d(hidden_size, hidden_size // 2, (2, 1))
        self.conv2 = nn.Conv2d(hidden_size // 2, hidden_size, (2, 1))
        self.output_layernorm = YuanRMSNorm(hidden_size)

    def _conv_chain(self, x):
        """x: [B, S+pad, C] already carrying the causal left context;
        returns y2 aligned with x[:, pad:]."""
        z = x.transpose(1, 2).unsqueeze(-1)            # [B, C, S', 1]
        y1 = self.conv1(z)                             # [B, C/2, S'-1, 1]
        y2 = self.conv2(y1)                            # [B, C, S'-2, 1]
        return y2.squeeze(-1).transpose(1, 2)          # [B, S'-2, C]

    def forward(self, x, before_hidden_states=None):
        B, S, C = x.shape
        if before_hidden_states is None:
            ctx = x.new_zeros(B, 2, C)                 # causal zero left-pad
        else:
            ctx = before_hidden_states                 # last 2 raw hiddens
        y2 = self._conv_chain(torch.cat([ctx, x], dim=1))
        return self.output_layernorm(y2 + x)


class YuanAttention(nn.Module):
    def __init__(self, config: YuanConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.num_kv_heads = config.num_key_value_heads
        self.head_dim = config.head_dim
        self.rope_theta = config.rope_theta
        kv = self.num_kv_heads * self.head_dim
        self.q_proj = nn.Linear(h, h, bias=False)
        self.k_proj = nn.Linear(h, kv, bias=False)
        self.v_proj = nn.Linear(h, kv, bias=False)
        self.o_proj = nn.Linear(h, h, bias=False)
        self.lf_gate = LocalizedFiltering(h)

    def _cos_sin(self, S, device, offset):
        inv = 1.0 / (self.rope_theta ** (
            torch.arange(0, self.head_dim, 2, device=device).float() / self.head_dim))
        t = torch.arange(offset, offset + S, device=device).float()
        freqs = torch.outer(t, inv)
        emb = torch.cat([freqs, freqs], dim=-1)
        return emb.cos(), emb.sin()

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        B, S, H = x.shape
        # V reads the raw hidden states
        v = self.v_proj(x).view(B, S, self.num_kv_heads, self.head_dim)
        # Q/K read the localized-filtered states; memory = last 2 raw hiddens
        before = past_key_value[2] if past_key_value is not None else None
        lf = self.lf_gate(x, before)
        if use_cache:
            if S >= 2:
                memory = x[:, -2:, :]
            else:
                prev = (before[:, -1:, :] if before is not None
                        else torch.zeros_like(x[:, :1]))
                memory = torch.cat([prev, x], dim=1)
        q = self.q_proj(lf).view(B, S, self.num_heads, self.head_dim)
        k = self.k_proj(lf).view(B, S, self.num_kv_heads, self.head_dim)
        cos, sin = self._cos_sin(S, x.device, position_offset)
        q, k = ops.fused_rope(q, k, cos.to(x.dtype), sin.to(x.dtype))
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v, memory) if use_cache else None
        out = ops.flash_attention(q, k, v, causal=True)
        out = self.o_proj(out.reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class YuanMLP(nn.Module):
    def __init__(self, config: YuanConfig):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.gate_proj = nn.Linear(h, i, bias=False)
        self.up_proj = nn.Linear(h, i, bias=False)
        self.down_proj = nn.Linear(i, h, bias=False)

    def forward(self, x):
        # reference yuan:41 — silu on UP, multiplied by GATE
        return self.down_proj(ops.swiglu(self.up_proj(x), self.gate_proj(x)))


class YuanDecoderLayer(nn.Module):
    def __init__(self, config: YuanConfig):
        super().__init__()
        self.input_layernorm = YuanRMSNorm(config.hidden_size, config.rms_norm_eps)
        self.self_attn = YuanAttention(config)
        self.post_attention_layernorm = YuanRMSNorm(config.hidden_size,
                                                    config.rms_norm_eps)
        self.mlp = YuanMLP(config)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        attn = self.self_attn(self.input_layernorm(x), past_key_value,
                              use_cache, position_offset)
        if use_cache:
            attn, present = attn
        x = x + attn
        x = x + self.mlp(self.post_attention_layernorm(x))
        if use_cache:
            return x, present
        return x


class YuanPretrainedModel(PretrainedModel):
    config_class = YuanConfig
    base_model_prefix = "yuan"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, YuanRMSNorm):
            module.weight.data.fill_(1.0)


class YuanModel(YuanPretrainedModel):
    def __init__(self, config: YuanConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [YuanDecoderLayer(config) for _ in range(config.num_hidden_layers)])
        self.norm = YuanRMSNorm(config.hidden_size, config.rms_norm_eps)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embed_tokens(input_ids)
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, past, use_cache, offset)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.norm(x)
        if use_cache:
            return x, presents
        return x


class YuanForCausalLM(YuanPretrainedModel, GenerationMixin):
    def __init__(self, config: YuanConfig):
        super().__init__(config)
        self.yuan = YuanModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.yuan.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.yuan(input_ids, past_key_values, use_cache)
        if use_cache:
            hidden, presents = out
        else:
            hidden, presents = out, None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                -100, reduction="mean")
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
