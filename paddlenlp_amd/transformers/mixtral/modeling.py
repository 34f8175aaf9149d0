"""Mixtral (sparse MoE) model family.

Reference behavior: paddlenlp/transformers/mixtral/modeling.py —
MixtralSparseMoeBlock :490 (top-k router + expert MLPs),
load_balancing_loss_func :67 (aux loss over router probs).  The attention
path reuses the Llama GQA attention (same kernels); the MoE block routes
tokens through parallel.expert_parallel's all-to-all when
expert_parallel_degree > 1, with expert params flagged no_sync so the DP
gradient all-reduce skips them (reference trainer.py:1079-1085).
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
from ..model_utils import PretrainedModel
from ..llama.modeling import LlamaAttention, LlamaRMSNorm
from .configuration import MixtralConfig

__all__ = ["MixtralModel", "MixtralForCausalLM", "MixtralSparseMoeBlock",
           "load_balancing_loss_func"]


def load_balancing_loss_func(router_logits_list, num_experts: int, top_k: int):
    """Switch-style aux loss (reference mixtral/modeling.py:67)."""
    if not router_logits_list:
        return torch.zeros(())
    logits = torch.cat([l.reshape(-1, num_experts) for l in router_logits_list], dim=0)
    probs = logits.float().softmax(-1)
    _, selected = probs.topk(top_k, dim=-1)
    mask = F.one_hot(selected, num_experts).float().max(dim=1).values  # [T, E]
    tokens_per_expert = mask.mean(0)
    router_prob_per_expert = probs.mean(0)
    return (tokens_per_expert * router_prob_per_expert).sum() * num_experts


class MixtralExpertMLP(nn.Module):
    def __init__(self, config: MixtralConfig):
        super().__init__()
        h, i = config.hidden_size, config.intermediate_size
        self.w1 = nn.Linear(h, i, bias=False)   # gate
        self.w3 = nn.Linear(h, i, bias=False)   # up
        self.w2 = nn.Linear(i, h, bias=False)   # down

    def forward(self, x):
        return self.w2(ops.swiglu(torch.cat([self.w1(x), self.w3(x)], dim=-1)))


class MixtralSparseMoeBlock(nn.Module):
    def __init__(self, config: MixtralConfig):
        super().__init__()
        self.config = config
        self.num_experts = config.num_local_experts
        self.top_k = config.num_experts_per_tok
        self.gate = nn.Linear(config.hidden_size, self.num_experts, bias=False)

        topo = get_topology()
        self.ep_degree = config.expert_parallel_degree
        self.ep_group = None
        if self.ep_degree > 1:
            # EP shares the dp axis (reference: topo reorders so dp is
            # innermost under use_expert_parallel, training_args.py:1275-76)
            self.ep_group = topo.data_parallel_group
            assert self.num_experts % self.ep_degree == 0
            self.experts_per_rank = self.num_experts // self.ep_degree
        else:
            self.experts_per_rank = self.num_experts
        # stacked weights -> one batched GEMM per projection
        # (parallel.expert_parallel.GroupedExperts; per-expert state-dict
        # keys are preserved)
        self.experts = GroupedExperts(
            self.experts_per_rank, config.hidden_size, config.intermediate_size)
        if self.ep_degree > 1:
            mark_moe_params_no_sync(self.experts)

    def forward(self, hidden):  # [B, S, H]
        B, S, H = hidden.shape
        x = hidden.reshape(-1, H)
        router_logits = self.gate(x)  # [T, E]
        probs = router_logits.float().softmax(-1)
        topk_w, topk_e = probs.topk(self.top_k, dim=-1)
        topk_w = topk_w / topk_w.sum(-1, keepdim=True)
        topk_w = topk_w.to(hidden.dtype)

        # repeat tokens per top-k slot and dispatch
        T = x.shape[0]
        flat_x = x.repeat_interleave(self.top_k, dim=0)          # [T*k, H]
        flat_e = topk_e.reshape(-1)                               # [T*k]
        out_flat = dispatch_and_combine(
            flat_x, flat_e, self.num_experts,
            grouped_fn=self.experts.forward_grouped,
            group=self.ep_group,
        )
        out = (out_flat.reshape(T, self.top_k, H) * topk_w[..., None]).sum(1)
        return out.reshape(B, S, H), router_logits


class MixtralDecoderLayer(nn.Module):
    def __init__(self, config: MixtralConfig, layer_idx: int = 0):
        super().__init__()
        self.self_attn = LlamaAttention(config, layer_idx)
        self.block_sparse_moe = MixtralSparseMoeBlock(config)
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
        moe_out, router_logits = self.block_sparse_moe(self.post_attention_layernorm(x))
        x = residual + moe_out
        if use_cache:
            return x, present, router_logits
        return x, router_logits


class MixtralPretrainedModel(PretrainedModel):
    config_class = MixtralConfig
    base_model_prefix = "mixtral"


class MixtralModel(MixtralPretrainedModel):
    def __init__(self, config: MixtralConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [MixtralDecoderLayer(config, i) for i in range(config.num_hidden_layers)]
        )
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
            if self.config.recompute and self.training and past is None:
                x, router_logits = checkpoint(layer, x, None, False, offset, use_reentrant=False)
            else:
                out = layer(x, past, use_cache, offset)
                if use_cache:
                    x, present, router_logits = out
                    presents.append(present)
                else:
                    x, router_logits = out
            all_router_logits.append(router_logits)
        x = self.norm(x)
        if use_cache:
            return x, presents, all_router_logits
        return x, all_router_logits


class MixtralForCausalLM(MixtralPretrainedModel, GenerationMixin):
    def __init__(self, config: MixtralConfig):
        super().__init__(config)
        self.mixtral = MixtralModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.mixtral.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.mixtral(input_ids, past_key_values, use_cache)
        if use_cache:
            hidden, presents, router_logits = out
        else:
            hidden, router_logits = out
            presents = None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                -100, reduction="mean",
            )
            aux = load_balancing_loss_func(
                router_logits, self.config.num_local_experts,
                self.config.num_experts_per_tok,
            ).to(loss.device)
            loss = loss + self.config.router_aux_loss_coef * aux
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
