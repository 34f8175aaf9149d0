"""QWen v1 family (reference: paddlenlp/transformers/qwen/modeling.py).

Llama-like decoder with QWen's quirks: fused ``c_attn`` qkv projection that
ALWAYS carries bias (all other linears follow ``no_bias``), split-half
SwiGLU MLP (``c_proj(w1(x) * silu(w2(x)))`` with ff = intermediate//2),
RMSNorm (ln_1/ln_2/ln_f), full-head NeoX rotary, optional logn attention
scaling for inference beyond ``seq_length``, untied LM head.  Runs on the
CDNA4 fused ops (flash attention [B,S,H,D], fused rope, rms_norm, swiglu).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..model_utils import PretrainedModel
from .configuration import QWenConfig

__all__ = ["QWenModel", "QWenForCausalLM", "QWenLMHeadModel"]


class QWenRMSNorm(nn.Module):
    def __init__(self, hidden_size, eps=1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(hidden_size))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


class QWenAttention(nn.Module):
    def __init__(self, config: QWenConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.rope_base = config.rotary_emb_base
        self.use_logn_attn = config.use_logn_attn
        self.seq_length = config.seq_length
        # c_attn always has bias; c_proj follows no_bias
        self.c_attn = nn.Linear(h, 3 * h, bias=True)
        self.c_proj = nn.Linear(h, h, bias=not config.no_bias)

    def _cos_sin(self, S, device, offset):
        inv = 1.0 / (self.rope_base ** (
            torch.arange(0, self.head_dim, 2, device=device).float() / self.head_dim))
        t = torch.arange(offset, offset + S, device=device).float()
        freqs = torch.outer(t, inv)
        emb = torch.cat([freqs, freqs], dim=-1)
        return emb.cos(), emb.sin()

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        B, S, H = x.shape
        qkv = self.c_attn(x)
        q, k, v = qkv.chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        q, k, v = q.view(shape), k.view(shape), v.view(shape)
        cos, sin = self._cos_sin(S, x.device, position_offset)
        q, k = ops.fused_rope(q, k, cos.to(x.dtype), sin.to(x.dtype))
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v) if use_cache else None
        if self.use_logn_attn and not self.training:
            # scale queries at absolute positions beyond the training length
            pos = torch.arange(position_offset, position_offset + S,
                               device=x.device, dtype=torch.float32)
            logn = torch.where(pos + 1 > self.seq_length,
                               torch.log(pos + 1) / math.log(self.seq_length),
                               torch.ones_like(pos))
            q = q * logn.to(q.dtype)[None, :, None, None]
        out = ops.flash_attention(q, k, v, causal=True)
        out = self.c_proj(out.reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class QWenMLP(nn.Module):
    """c_proj(w1(x) * silu(w2(x))); ff = intermediate_size // 2."""

    def __init__(self, config: QWenConfig):
        super().__init__()
        h = config.hidden_size
        ff = config.intermediate_size // 2
        bias = not config.no_bias
        self.w1 = nn.Linear(h, ff, bias=bias)
        self.w2 = nn.Linear(h, ff, bias=bias)
        self.c_proj = nn.Linear(ff, h, bias=bias)

    def forward(self, x):
        # swiglu(cat[gate, up]) = silu(gate) * up with gate=w2, up=w1
        return self.c_proj(ops.swiglu(self.w2(x), self.w1(x)))


class QWenBlock(nn.Module):
    def __init__(self, config: QWenConfig):
        super().__init__()
        self.ln_1 = QWenRMSNorm(config.hidden_size, config.layer_norm_epsilon)
        self.attn = QWenAttention(config)
        self.ln_2 = QWenRMSNorm(config.hidden_size, config.layer_norm_epsilon)
        self.mlp = QWenMLP(config)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        attn = self.attn(self.ln_1(x), past_key_value, use_cache, position_offset)
        if use_cache:
            attn, present = attn
        x = x + attn
        x = x + self.mlp(self.ln_2(x))
        if use_cache:
            return x, present
        return x


class QWenPretrainedModel(PretrainedModel):
    config_class = QWenConfig
    base_model_prefix = "qwen"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, QWenRMSNorm):
            module.weight.data.fill_(1.0)


class QWenModel(QWenPretrainedModel):
    def __init__(self, config: QWenConfig):
        super().__init__(config)
        self.wte = nn.Embedding(config.vocab_size, config.hidden_size)
        self.h = nn.ModuleList(
            [QWenBlock(config) for _ in range(config.num_hidden_layers)])
        self.ln_f = QWenRMSNorm(config.hidden_size, config.layer_norm_epsilon)
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


class QWenForCausalLM(QWenPretrainedModel, GenerationMixin):
    def __init__(self, config: QWenConfig):
        super().__init__(config)
        self.qwen = QWenModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.qwen.wte

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.qwen(input_ids, past_key_values, use_cache)
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


# reference exposes the HF-style alias
QWenLMHeadModel = QWenForCausalLM
