"""Gemma model family (reference: paddlenlp/transformers/gemma/modeling.py).

Llama-architecture decoder with Gemma specifics: RMSNorm computes
x_hat * (1 + weight) with zero-initialized weight, embeddings scaled by
sqrt(hidden_size), GELU(tanh) gated MLP, explicit head_dim decoupled from
hidden/heads, tied LM head.  Attention reuses the Llama GQA path (same
CDNA4 flash kernels, [B, S, H, D] layout).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..llama.modeling import LlamaAttention
from ..model_utils import PretrainedModel
from .configuration import GemmaConfig

__all__ = ["GemmaModel", "GemmaForCausalLM"]


class GemmaRMSNorm(nn.Module):
    """x_hat * (1 + w); w zero-init (reference GemmaRMSNorm)."""

    def __init__(self, config: GemmaConfig):
        super().__init__()
        self.weight = nn.Parameter(torch.zeros(config.hidden_size))
        self.eps = config.rms_norm_eps

    def forward(self, x):
        xf = x.float()
        xhat = xf * torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + self.eps)
        return (xhat * (1.0 + self.weight.float())).to(x.dtype)


class GemmaMLP(nn.Module):
    def __init__(self, config: GemmaConfig):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.gate_proj = nn.Linear(h, i, bias=False)
        self.up_proj = nn.Linear(h, i, bias=False)
        self.down_proj = nn.Linear(i, h, bias=False)

    def forward(self, x):
        return self.down_proj(
            F.gelu(self.gate_proj(x), approximate="tanh") * self.up_proj(x))


class GemmaDecoderLayer(nn.Module):
    def __init__(self, config: GemmaConfig, layer_idx: int = 0):
        super().__init__()
        self.self_attn = LlamaAttention(config, layer_idx)
        self.mlp = GemmaMLP(config)
        self.input_layernorm = GemmaRMSNorm(config)
        self.post_attention_layernorm = GemmaRMSNorm(config)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        residual = x
        attn = self.self_attn(self.input_layernorm(x), None, None,
                              past_key_value, use_cache, position_offset)
        if use_cache:
            attn, present = attn
        x = residual + attn
        x = x + self.mlp(self.post_attention_layernorm(x))
        if use_cache:
            return x, present
        return x


class GemmaPretrainedModel(PretrainedModel):
    config_class = GemmaConfig
    base_model_prefix = "gemma"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, GemmaRMSNorm):
            module.weight.data.zero_()  # (1 + w) form


class GemmaModel(GemmaPretrainedModel):
    def __init__(self, config: GemmaConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [GemmaDecoderLayer(config, i)
             for i in range(config.num_hidden_layers)])
        self.norm = GemmaRMSNorm(config)
        self.embed_scale = math.sqrt(config.hidden_size)
        # scaled embeddings x tied head blow up under torch's default
        # N(0,1) embedding init: apply the family init at construction
        self.init_weights()

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embed_tokens(input_ids) * self.embed_scale
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            if getattr(self.config, "recompute", False) and self.training and past is None:
                x = checkpoint(layer, x, None, False, offset, use_reentrant=False)
            else:
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


class GemmaForCausalLM(GemmaPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: GemmaConfig):
        super().__init__(config)
        self.gemma = GemmaModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.gemma.embed_tokens.weight
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.gemma.embed_tokens.weight

    def get_input_embeddings(self):
        return self.gemma.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.gemma(input_ids, past_key_values, use_cache)
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
