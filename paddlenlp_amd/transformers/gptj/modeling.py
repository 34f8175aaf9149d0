"""GPT-J model family (reference: paddlenlp/transformers/gptj/modeling.py).

Parallel-residual decoder (attention and MLP read the same layernormed
input, falcon-style) with rotary position embedding on only the first
`rotary_dim` dims of each head in interleaved (even, odd) pairs, no-bias
attention projections, GELU MLP, untied LM head with bias.  Attention runs
through the flash-attention seam ([B, S, H, D]).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..model_utils import PretrainedModel
from .configuration import GPTJConfig

__all__ = ["GPTJModel", "GPTJForCausalLM"]


def _gptj_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor,
               rotary_dim: int):
    """Interleaved-pair rotary on the first rotary_dim head dims.

    x: [B, S, H, D]; cos/sin: [S, rotary_dim//2]."""
    rot, passthrough = x[..., :rotary_dim], x[..., rotary_dim:]
    x0 = rot[..., 0::2]
    x1 = rot[..., 1::2]
    c = cos[None, :, None, :]
    s = sin[None, :, None, :]
    out = torch.stack([x0 * c - x1 * s, x1 * c + x0 * s], dim=-1).flatten(-2)
    return torch.cat([out, passthrough], dim=-1)


class GPTJAttention(nn.Module):
    def __init__(self, config: GPTJConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.rotary_dim = config.rotary_dim
        self.rope_theta = config.rope_theta
        self.q_proj = nn.Linear(h, h, bias=False)
        self.k_proj = nn.Linear(h, h, bias=False)
        self.v_proj = nn.Linear(h, h, bias=False)
        self.out_proj = nn.Linear(h, h, bias=False)

    def _cos_sin(self, S, device, dtype, offset):
        n = self.rotary_dim // 2
        inv = 1.0 / (self.rope_theta
                     ** (torch.arange(n, device=device).float() * 2 / self.rotary_dim))
        t = torch.arange(offset, offset + S, device=device).float()
        freqs = torch.outer(t, inv)
        return freqs.cos().to(dtype), freqs.sin().to(dtype)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        B, S, H = x.shape
        shape = (B, S, self.num_heads, self.head_dim)
        q = self.q_proj(x).view(shape)
        k = self.k_proj(x).view(shape)
        v = self.v_proj(x).view(shape)
        cos, sin = self._cos_sin(S, x.device, x.dtype, position_offset)
        q = _gptj_rope(q, cos, sin, self.rotary_dim)
        k = _gptj_rope(k, cos, sin, self.rotary_dim)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v) if use_cache else None
        out = ops.flash_attention(q, k, v, causal=True)
        out = self.out_proj(out.reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class GPTJBlock(nn.Module):
    def __init__(self, config: GPTJConfig):
        super().__init__()
        h = config.hidden_size
        self.ln_1 = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.attn = GPTJAttention(config)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        ln = self.ln_1(x)
        attn = self.attn(ln, past_key_value, use_cache, position_offset)
        if use_cache:
            attn, present = attn
        # parallel residual: x + attn(ln) + mlp(ln)
        mlp = self.fc_out(F.gelu(self.fc_in(ln), approximate="tanh"))
        x = x + attn + mlp
        if use_cache:
            return x, present
        return x


class GPTJPretrainedModel(PretrainedModel):
    config_class = GPTJConfig
    base_model_prefix = "gptj"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, nn.LayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()


class GPTJModel(GPTJPretrainedModel):
    def __init__(self, config: GPTJConfig):
        super().__init__(config)
        self.wte = nn.Embedding(config.vocab_size, config.hidden_size)
        self.h = nn.ModuleList(
            [GPTJBlock(config) for _ in range(config.num_hidden_layers)])
        self.ln_f = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.wte

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.wte(input_ids)
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        presents = [] if use_cache else None
        for i, block in enumerate(self.h):
            past = past_key_values[i] if past_key_values is not None else None
            out = block(x, past, use_cache, offset)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.ln_f(x)
        if use_cache:
            return x, presents
        return x


class GPTJForCausalLM(GPTJPretrainedModel, GenerationMixin):
    def __init__(self, config: GPTJConfig):
        super().__init__(config)
        self.gptj = GPTJModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=True)
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.gptj.wte

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.gptj(input_ids, past_key_values, use_cache)
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
