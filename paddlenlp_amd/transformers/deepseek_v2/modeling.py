"""DeepSeek-V2 model family (reference: paddlenlp/transformers/deepseek_v2/
modeling.py — MLA attention, grouped top-k MoE router).

MLA (multi-head latent attention): queries and keys/values are projected
through low-rank latents (q_lora_rank / kv_lora_rank) and split into a
non-positional part plus a SHARED rotary part broadcast across heads (the
latent is what a serving cache would store).  qk_head_dim (192) differs from
v_head_dim (128), so attention runs through SDPA rather than the D-uniform
flash kernel; a latent-absorbed decode kernel is the planned serving path.

DeepseekMoE: optional shared experts + routed experts with group-limited
top-k (scores are grouped over n_group expert groups; only the topk_group
best groups compete), scaled by routed_scaling_factor.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ...parallel.expert_parallel import dispatch_and_combine
from ..llama.modeling import LlamaRMSNorm
from ..mixtral.modeling import load_balancing_loss_func
from ..model_utils import PretrainedModel
from .configuration import DeepseekV2Config

__all__ = ["DeepseekV2Model", "DeepseekV2ForCausalLM", "DeepseekV2Attention",
           "DeepseekV2MoEGate"]


def _rope_cos_sin(seq_len, dim, theta, device, dtype, offset=0):
    inv = 1.0 / (theta ** (torch.arange(0, dim, 2, device=device).float() / dim))
    t = torch.arange(offset, offset + seq_len, device=device).float()
    freqs = torch.outer(t, inv)                      # [S, dim/2]
    emb = torch.cat([freqs, freqs], dim=-1)
    return emb.cos().to(dtype), emb.sin().to(dtype)


def _rotate_half(x):
    half = x.shape[-1] // 2
    return torch.cat([-x[..., half:], x[..., :half]], dim=-1)


def _apply_rope(x, cos, sin):
    """x: [B, H, S, D_rope] (interleaved pair layout folded as halves)."""
    return x * cos + _rotate_half(x) * sin


class DeepseekV2Attention(nn.Module):
    def __init__(self, config: DeepseekV2Config):
        super().__init__()
        c = config
        h = c.hidden_size
        self.num_heads = c.num_attention_heads
        self.qk_nope = c.qk_nope_head_dim
        self.qk_rope = c.qk_rope_head_dim
        self.qk_dim = c.qk_head_dim
        self.v_dim = c.v_head_dim
        self.softmax_scale = self.qk_dim ** -0.5
        self.rope_theta = c.rope_theta

        if c.q_lora_rank:
            self.q_a_proj = nn.Linear(h, c.q_lora_rank, bias=False)
            self.q_a_layernorm = LlamaRMSNorm(c, c.q_lora_rank)
            self.q_b_proj = nn.Linear(c.q_lora_rank,
                                      self.num_heads * self.qk_dim, bias=False)
        else:
            self.q_proj = nn.Linear(h, self.num_heads * self.qk_dim, bias=False)
        # kv latent + the shared rope key, in one projection
        self.kv_a_proj_with_mqa = nn.Linear(
            h, c.kv_lora_rank + self.qk_rope, bias=False)
        self.kv_a_layernorm = LlamaRMSNorm(c, c.kv_lora_rank)
        self.kv_b_proj = nn.Linear(
            c.kv_lora_rank, self.num_heads * (self.qk_nope + self.v_dim), bias=False)
        self.o_proj = nn.Linear(self.num_heads * self.v_dim, h, bias=False)
        self.kv_lora_rank = c.kv_lora_rank

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        B, S, _ = x.shape
        H = self.num_heads
        if hasattr(self, "q_a_proj"):
            q = self.q_b_proj(self.q_a_layernorm(self.q_a_proj(x)))
        else:
            q = self.q_proj(x)
        q = q.view(B, S, H, self.qk_dim).transpose(1, 2)      # [B,H,S,Dqk]
        q_nope, q_rope = q.split([self.qk_nope, self.qk_rope], dim=-1)

        kv_a = self.kv_a_proj_with_mqa(x)                      # [B,S,rank+rope]
        latent, k_rope = kv_a.split([self.kv_lora_rank, self.qk_rope], dim=-1)
        kv = self.kv_b_proj(self.kv_a_layernorm(latent))
        kv = kv.view(B, S, H, self.qk_nope + self.v_dim).transpose(1, 2)
        k_nope, v = kv.split([self.qk_nope, self.v_dim], dim=-1)
        k_rope = k_rope.view(B, S, 1, self.qk_rope).transpose(1, 2)  # shared

        cos, sin = _rope_cos_sin(S, self.qk_rope, self.rope_theta,
                                 x.device, x.dtype, position_offset)
        q_rope = _apply_rope(q_rope, cos, sin)
        k_rope = _apply_rope(k_rope, cos, sin)

        k = torch.cat([k_nope, k_rope.expand(B, H, S, self.qk_rope)], dim=-1)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=2)
            v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None

        Skv = k.shape[2]
        is_causal = S > 1 and S == Skv
        attn_mask = None
        if S > 1 and S != Skv:
            m = torch.full((S, Skv), float("-inf"), device=x.device, dtype=x.dtype)
            attn_mask = m.triu(Skv - S + 1)
        out = F.scaled_dot_product_attention(
            q, k, v, attn_mask=attn_mask, is_causal=is_causal,
            scale=self.softmax_scale)
        out = out.transpose(1, 2).reshape(B, S, H * self.v_dim)
        out = self.o_proj(out)
        if use_cache:
            return out, present
        return out


class DeepseekV2MLP(nn.Module):
    def __init__(self, config: DeepseekV2Config, intermediate_size=None):
        super().__init__()
        h = config.hidden_size
        i = intermediate_size or config.intermediate_size
        self.gate_proj = nn.Linear(h, i, bias=False)
        self.up_proj = nn.Linear(h, i, bias=False)
        self.down_proj = nn.Linear(i, h, bias=False)

    def forward(self, x):
        return self.down_proj(
            ops.swiglu(torch.cat([self.gate_proj(x), self.up_proj(x)], dim=-1)))


class DeepseekV2MoEGate(nn.Module):
    """Group-limited top-k routing (reference MoEGate): softmax scores are
    bucketed into n_group expert groups; only experts inside the topk_group
    best groups (by per-group max score) are candidates."""

    def __init__(self, config: DeepseekV2Config):
        super().__init__()
        self.config = config
        self.weight = nn.Parameter(
            torch.empty(config.n_routed_experts, config.hidden_size))
        nn.init.kaiming_uniform_(self.weight, a=math.sqrt(5))

    def forward(self, x):  # x: [T, H]
        c = self.config
        logits = F.linear(x.float(), self.weight.float())
        scores = logits.softmax(-1)                      # [T, E]
        if c.n_group > 1:
            T = scores.shape[0]
            grouped = scores.view(T, c.n_group, -1)
            group_scores = grouped.amax(-1)              # [T, G]
            top_groups = group_scores.topk(c.topk_group, dim=-1).indices
            mask = torch.zeros_like(group_scores)
            mask.scatter_(1, top_groups, 1.0)
            scores = (grouped * mask[..., None]).view(T, -1)
        topk_w, topk_e = scores.topk(c.num_experts_per_tok, dim=-1)
        if c.norm_topk_prob:
            topk_w = topk_w / topk_w.sum(-1, keepdim=True).clamp(min=1e-20)
        return topk_w * c.routed_scaling_factor, topk_e, logits


class DeepseekV2MoE(nn.Module):
    def __init__(self, config: DeepseekV2Config):
        super().__init__()
        self.config = config
        self.gate = DeepseekV2MoEGate(config)
        self.experts = nn.ModuleList(
            [DeepseekV2MLP(config, config.moe_intermediate_size)
             for _ in range(config.n_routed_experts)])
        self.shared_experts = None
        if config.n_shared_experts:
            self.shared_experts = DeepseekV2MLP(
                config, config.moe_intermediate_size * config.n_shared_experts)

    def forward(self, hidden):
        B, S, H = hidden.shape
        x = hidden.reshape(-1, H)
        topk_w, topk_e, router_logits = self.gate(x)
        k = topk_e.shape[1]
        flat_x = x.repeat_interleave(k, dim=0)
        flat_e = topk_e.reshape(-1)
        out_flat = dispatch_and_combine(
            flat_x, flat_e, self.config.n_routed_experts,
            expert_fn=lambda le, toks: self.experts[le](toks), group=None)
        out = (out_flat.reshape(-1, k, H) * topk_w[..., None].to(hidden.dtype)).sum(1)
        if self.shared_experts is not None:
            out = out + self.shared_experts(x)
        return out.reshape(B, S, H), router_logits


def _use_moe(config: DeepseekV2Config, layer_idx: int) -> bool:
    return (config.n_routed_experts is not None
            and layer_idx >= config.first_k_dense_replace
            and layer_idx % config.moe_layer_freq == 0)


class DeepseekV2DecoderLayer(nn.Module):
    def __init__(self, config: DeepseekV2Config, layer_idx: int):
        super().__init__()
        self.self_attn = DeepseekV2Attention(config)
        self.mlp = (DeepseekV2MoE(config) if _use_moe(config, layer_idx)
                    else DeepseekV2MLP(config))
        self.is_moe = isinstance(self.mlp, DeepseekV2MoE)
        self.input_layernorm = LlamaRMSNorm(config)
        self.post_attention_layernorm = LlamaRMSNorm(config)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        residual = x
        attn = self.self_attn(self.input_layernorm(x), past_key_value,
                              use_cache, position_offset)
        if use_cache:
            attn, present = attn
        x = residual + attn
        residual = x
        h = self.post_attention_layernorm(x)
        if self.is_moe:
            mlp_out, router_logits = self.mlp(h)
        else:
            mlp_out, router_logits = self.mlp(h), None
        x = residual + mlp_out
        if use_cache:
            return x, present, router_logits
        return x, router_logits


class DeepseekV2PretrainedModel(PretrainedModel):
    config_class = DeepseekV2Config
    base_model_prefix = "deepseek_v2"


class DeepseekV2Model(DeepseekV2PretrainedModel):
    def __init__(self, config: DeepseekV2Config):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [DeepseekV2DecoderLayer(config, i)
             for i in range(config.num_hidden_layers)])
        self.norm = LlamaRMSNorm(config)

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embed_tokens(input_ids)
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[2]
        presents = [] if use_cache else None
        all_router_logits = []
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, past, use_cache, offset)
            if use_cache:
                x, present, router_logits = out
                presents.append(present)
            else:
                x, router_logits = out
            if router_logits is not None:
                all_router_logits.append(router_logits)
        x = self.norm(x)
        if use_cache:
            return x, presents, all_router_logits
        return x, all_router_logits


class DeepseekV2ForCausalLM(DeepseekV2PretrainedModel, GenerationMixin):
    def __init__(self, config: DeepseekV2Config):
        super().__init__(config)
        self.deepseek_v2 = DeepseekV2Model(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.deepseek_v2.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.deepseek_v2(input_ids, past_key_values, use_cache)
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
            if router_logits and self.config.aux_loss_alpha > 0:
                aux = load_balancing_loss_func(
                    router_logits, self.config.n_routed_experts,
                    self.config.num_experts_per_tok).to(loss.device)
                loss = loss + self.config.aux_loss_alpha * aux
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
