"""Jamba model family (reference: paddlenlp/transformers/jamba/modeling.py).

Hybrid decoder: most layers are Mamba selective-SSM mixers (position comes
from the recurrence — attention layers carry NO rope), every
attn_layer_period-th layer (at attn_layer_offset) is GQA attention, and
every expert_layer_period-th layer (at expert_layer_offset) swaps the dense
MLP for a top-k MoE.  Composed from the existing MambaMixer, the Llama GQA
attention (rope disabled) and the Mixtral-style expert dispatch.
"""
from __future__ import annotations

from types import SimpleNamespace

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ...parallel.expert_parallel import dispatch_and_combine
from ..llama.modeling import LlamaRMSNorm
from ..mamba.modeling import MambaMixer
from ..mixtral.modeling import load_balancing_loss_func
from ..model_utils import PretrainedModel
from .configuration import JambaConfig

__all__ = ["JambaModel", "JambaForCausalLM"]


def _mamba_view(c: JambaConfig):
    """Adapter config for the reused MambaMixer (its intermediate_size is
    the SSM inner width, not the MLP width)."""
    return SimpleNamespace(
        hidden_size=c.hidden_size,
        intermediate_size=c.mamba_expand * c.hidden_size,
        state_size=c.mamba_d_state, conv_kernel=c.mamba_d_conv,
        time_step_rank=c.mamba_dt_rank, use_bias=False, use_conv_bias=True)


class JambaAttention(nn.Module):
    """GQA attention WITHOUT rotary embedding (reference JambaAttention —
    position information comes from the mamba layers)."""

    def __init__(self, config: JambaConfig):
        super().__init__()
        c = config
        self.num_heads = c.num_attention_heads
        self.num_kv_heads = c.num_key_value_heads
        self.head_dim = c.head_dim
        q_out = self.num_heads * self.head_dim
        kv_out = self.num_kv_heads * self.head_dim
        self.qkv_proj = nn.Linear(c.hidden_size, q_out + 2 * kv_out, bias=False)
        self.o_proj = nn.Linear(q_out, c.hidden_size, bias=False)
        self.q_out, self.kv_out = q_out, kv_out

    def forward(self, x, cache=None):
        B, S, _ = x.shape
        q, k, v = self.qkv_proj(x).split(
            [self.q_out, self.kv_out, self.kv_out], dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim)
        k = k.view(B, S, self.num_kv_heads, self.head_dim)
        v = v.view(B, S, self.num_kv_heads, self.head_dim)
        if cache is not None:
            k = torch.cat([cache[0], k], dim=1)
            v = torch.cat([cache[1], v], dim=1)
        out = ops.flash_attention(q, k, v, causal=True)
        return self.o_proj(out.reshape(B, S, self.q_out)), (k, v)


class JambaMLP(nn.Module):
    def __init__(self, config: JambaConfig):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.gate_proj = nn.Linear(h, i, bias=False)
        self.up_proj = nn.Linear(h, i, bias=False)
        self.down_proj = nn.Linear(i, h, bias=False)

    def forward(self, x):
        return self.down_proj(
            ops.swiglu(torch.cat([self.gate_proj(x), self.up_proj(x)], dim=-1)))


class JambaMoE(nn.Module):
    def __init__(self, config: JambaConfig):
        super().__init__()
        self.num_experts = config.num_experts
        self.top_k = config.num_experts_per_tok
        self.router = nn.Linear(config.hidden_size, self.num_experts, bias=False)
        self.experts = nn.ModuleList(
            [JambaMLP(config) for _ in range(self.num_experts)])

    def forward(self, hidden):
        B, S, H = hidden.shape
        x = hidden.reshape(-1, H)
        router_logits = self.router(x)
        probs = router_logits.float().softmax(-1)
        topk_w, topk_e = probs.topk(self.top_k, dim=-1)
        topk_w = (topk_w / topk_w.sum(-1, keepdim=True)).to(hidden.dtype)
        flat_x = x.repeat_interleave(self.top_k, dim=0)
        out_flat = dispatch_and_combine(
            flat_x, topk_e.reshape(-1), self.num_experts,
            expert_fn=lambda le, toks: self.experts[le](toks), group=None)
        out = (out_flat.reshape(-1, self.top_k, H) * topk_w[..., None]).sum(1)
        return out.reshape(B, S, H), router_logits


class JambaLayer(nn.Module):
    def __init__(self, config: JambaConfig, layer_idx: int):
        super().__init__()
        c = config
        self.is_attn = (layer_idx % c.attn_layer_period == c.attn_layer_offset)
        self.is_moe = (layer_idx % c.expert_layer_period == c.expert_layer_offset)
        self.input_layernorm = LlamaRMSNorm(c)
        self.mixer = (JambaAttention(c) if self.is_attn
                      else MambaMixer(_mamba_view(c)))
        self.pre_ff_layernorm = LlamaRMSNorm(c)
        self.feed_forward = JambaMoE(c) if self.is_moe else JambaMLP(c)

    def forward(self, x, cache=None):
        h, new_cache = self.mixer(self.input_layernorm(x), cache)
        x = x + h
        ff = self.feed_forward(self.pre_ff_layernorm(x))
        router_logits = None
        if self.is_moe:
            ff, router_logits = ff
        x = x + ff
        return x, new_cache, router_logits


class JambaPretrainedModel(PretrainedModel):
    config_class = JambaConfig
    base_model_prefix = "jamba"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)


class JambaModel(JambaPretrainedModel):
    def __init__(self, config: JambaConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [JambaLayer(config, i) for i in range(config.num_hidden_layers)])
        self.final_layernorm = LlamaRMSNorm(config)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embed_tokens(input_ids)
        presents = [] if use_cache else None
        all_router_logits = []
        for i, layer in enumerate(self.layers):
            cache = past_key_values[i] if past_key_values is not None else None
            x, new_cache, router_logits = layer(x, cache)
            if use_cache:
                presents.append(new_cache)
            if router_logits is not None:
                all_router_logits.append(router_logits)
        x = self.final_layernorm(x)
        if use_cache:
            return x, presents, all_router_logits
        return x, all_router_logits


class JambaForCausalLM(JambaPretrainedModel, GenerationMixin):
    def __init__(self, config: JambaConfig):
        super().__init__(config)
        self.jamba = JambaModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.jamba.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.jamba(input_ids, past_key_values, use_cache)
        if use_cache:
            hidden, presents, router_logits = out
        else:
            hidden, router_logits = out
            presents = None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                -100, reduction="mean")
            if router_logits and self.config.router_aux_loss_coef > 0:
                aux = load_balancing_loss_func(
                    router_logits, self.config.num_experts,
                    self.config.num_experts_per_tok).to(loss.device)
                loss = loss + self.config.router_aux_loss_coef * aux
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
