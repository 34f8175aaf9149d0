"""Qwen2-MoE model family (reference: paddlenlp/transformers/qwen2_moe/modeling.py).

Mixtral-style top-k sparse MoE plus Qwen2 specifics: biased q/k/v
projections (via Qwen2Attention), a sigmoid-gated SHARED expert added to the
routed output, optional un-normalized top-k probs (norm_topk_prob=False),
and per-layer sparse/dense selection (decoder_sparse_step, mlp_only_layers).
EP routes through parallel.expert_parallel like Mixtral.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F
from torch.utils.checkpoint import checkpoint

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ...parallel.expert_parallel import (
    GroupedExperts,
    dispatch_and_combine,
    mark_moe_params_no_sync,
)
from ...parallel.topology import get_topology
from ..llama.modeling import LlamaRMSNorm
from ..mixtral.modeling import load_balancing_loss_func
from ..model_utils import PretrainedModel
from ..qwen2.modeling import Qwen2Attention
from .configuration import Qwen2MoeConfig

__all__ = ["Qwen2MoeModel", "Qwen2MoeForCausalLM", "Qwen2MoeSparseMoeBlock"]


class Qwen2MoeMLP(nn.Module):
    def __init__(self, config: Qwen2MoeConfig, intermediate_size: int):
        super().__init__()
        h = config.hidden_size
        self.gate_proj = nn.Linear(h, intermediate_size, bias=False)
        self.up_proj = nn.Linear(h, intermediate_size, bias=False)
        self.down_proj = nn.Linear(intermediate_size, h, bias=False)

    def forward(self, x):
        return self.down_proj(
            ops.swiglu(torch.cat([self.gate_proj(x), self.up_proj(x)], dim=-1)))


class Qwen2MoeSparseMoeBlock(nn.Module):
    """Routed experts + sigmoid-gated shared expert (reference
    Qwen2MoeSparseMoeBlock)."""

    def __init__(self, config: Qwen2MoeConfig):
        super().__init__()
        self.config = config
        self.num_experts = config.num_experts
        self.top_k = config.num_experts_per_tok
        self.norm_topk_prob = config.norm_topk_prob
        self.gate = nn.Linear(config.hidden_size, self.num_experts, bias=False)
        self.shared_expert = Qwen2MoeMLP(
            config, config.shared_expert_intermediate_size)
        self.shared_expert_gate = nn.Linear(config.hidden_size, 1, bias=False)

        topo = get_topology()
        self.ep_degree = config.expert_parallel_degree
        self.ep_group = None
        if self.ep_degree > 1:
            self.ep_group = topo.data_parallel_group
            assert self.num_experts % self.ep_degree == 0
            self.experts_per_rank = self.num_experts // self.ep_degree
        else:
            self.experts_per_rank = self.num_experts
        self.experts = GroupedExperts(
            self.experts_per_rank, config.hidden_size,
            config.moe_intermediate_size,
            names=("gate_proj", "up_proj", "down_proj"))
        if self.ep_degree > 1:
            mark_moe_params_no_sync(self.experts)

    def forward(self, hidden):  # [B, S, H]
        B, S, H = hidden.shape
        x = hidden.reshape(-1, H)
        router_logits = self.gate(x)
        probs = router_logits.float().softmax(-1)
        topk_w, topk_e = probs.topk(self.top_k, dim=-1)
        if self.norm_topk_prob:
            topk_w = topk_w / topk_w.sum(-1, keepdim=True)
        topk_w = topk_w.to(hidden.dtype)

        T = x.shape[0]
        flat_x = x.repeat_interleave(self.top_k, dim=0)
        flat_e = topk_e.reshape(-1)
        # the shared expert computes while the dispatch all-to-all is in
        # flight (reference overlaps fused_moe with shared-expert compute)
        shared_out = {}

        def _shared():
            shared_out["y"] = self.shared_expert(x) * torch.sigmoid(
                self.shared_expert_gate(x))

        out_flat = dispatch_and_combine(
            flat_x, flat_e, self.num_experts,
            grouped_fn=self.experts.forward_grouped,
            group=self.ep_group, overlap_fn=_shared)
        routed = (out_flat.reshape(T, self.top_k, H) * topk_w[..., None]).sum(1)
        out = routed + shared_out["y"]
        return out.reshape(B, S, H), router_logits


def _is_sparse_layer(config: Qwen2MoeConfig, layer_idx: int) -> bool:
    if layer_idx in config.mlp_only_layers:
        return False
    return (config.num_experts > 0
            and (layer_idx + 1) % config.decoder_sparse_step == 0)


class Qwen2MoeDecoderLayer(nn.Module):
    def __init__(self, config: Qwen2MoeConfig, layer_idx: int = 0):
        super().__init__()
        self.self_attn = Qwen2Attention(config, layer_idx)
        self.is_sparse = _is_sparse_layer(config, layer_idx)
        if self.is_sparse:
            self.mlp = Qwen2MoeSparseMoeBlock(config)
        else:
            self.mlp = Qwen2MoeMLP(config, config.intermediate_size)
        self.input_layernorm = LlamaRMSNorm(config)
        self.post_attention_layernorm = LlamaRMSNorm(config)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        residual = x
        h = self.input_layernorm(x)
        attn = self.self_attn(h, None, None, past_key_value, use_cache, position_offset)
        if use_cache:
            attn, present = attn
        x = residual + attn
        residual = x
        h = self.post_attention_layernorm(x)
        if self.is_sparse:
            mlp_out, router_logits = self.mlp(h)
        else:
            mlp_out, router_logits = self.mlp(h), None
        x = residual + mlp_out
        if use_cache:
            return x, present, router_logits
        return x, router_logits


class Qwen2MoePretrainedModel(PretrainedModel):
    config_class = Qwen2MoeConfig
    base_model_prefix = "qwen2_moe"


class Qwen2MoeModel(Qwen2MoePretrainedModel):
    def __init__(self, config: Qwen2MoeConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [Qwen2MoeDecoderLayer(config, i) for i in range(config.num_hidden_layers)])
        self.norm = LlamaRMSNorm(config)

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embed_tokens(input_ids)
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        presents = [] if use_cache else None
        all_router_logits = []
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            if getattr(self.config, "recompute", False) and self.training and past is None:
                x, router_logits = checkpoint(layer, x, None, False, offset,
                                              use_reentrant=False)
            else:
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


class Qwen2MoeForCausalLM(Qwen2MoePretrainedModel, GenerationMixin):
    def __init__(self, config: Qwen2MoeConfig):
        super().__init__(config)
        self.qwen2_moe = Qwen2MoeModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.qwen2_moe.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.qwen2_moe(input_ids, past_key_values, use_cache)
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
            aux = load_balancing_loss_func(
                router_logits, self.config.num_experts,
                self.config.num_experts_per_tok).to(loss.device)
            loss = loss + self.config.router_aux_loss_coef * aux
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
