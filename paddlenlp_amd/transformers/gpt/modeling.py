"""GPT-2/3 model family (reference: paddlenlp/transformers/gpt/modeling.py).

Pre-LN GPT-2: learned positional embeddings, LayerNorm, GELU MLP, fused QKV
with bias, tied LM head.  Attention runs through the same paddlenlp_amd.ops
flash-attention seam as Llama ([B, S, H, D] layout).
"""
from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..model_utils import PretrainedModel
from .configuration import GPTConfig

__all__ = ["GPTModel", "GPTForCausalLM", "GPTPretrainingCriterion"]


class GPTAttention(nn.Module):
    def __init__(self, config: GPTConfig):
        super().__init__()
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        h = config.hidden_size
        self.qkv_proj = nn.Linear(h, 3 * h, bias=True)
        self.out_proj = nn.Linear(h, h, bias=True)

    def forward(self, x, past_key_value=None, use_cache=False):
        B, S, H = x.shape
        qkv = self.qkv_proj(x)
        q, k, v = qkv.chunk(3, dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim)
        k = k.view(B, S, self.num_heads, self.head_dim)
        v = v.view(B, S, self.num_heads, self.head_dim)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v) if use_cache else None
        out = ops.flash_attention(q, k, v, causal=True)
        out = self.out_proj(out.reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class GPTMLP(nn.Module):
    def __init__(self, config: GPTConfig):
        super().__init__()
        self.fc_in = nn.Linear(config.hidden_size, config.intermediate_size, bias=True)
        self.fc_out = nn.Linear(config.intermediate_size, config.hidden_size, bias=True)

    def forward(self, x):
        return self.fc_out(F.gelu(self.fc_in(x), approximate="tanh"))


class GPTDecoderLayer(nn.Module):
    def __init__(self, config: GPTConfig):
        super().__init__()
        self.norm1 = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.self_attn = GPTAttention(config)
        self.norm2 = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.mlp = GPTMLP(config)

    def forward(self, x, past_key_value=None, use_cache=False):
        attn_out = self.self_attn(self.norm1(x), past_key_value, use_cache)
        if use_cache:
            attn_out, present = attn_out
        x = x + attn_out
        x = x + self.mlp(self.norm2(x))
        if use_cache:
            return x, present
        return x


class GPTPretrainedModel(PretrainedModel):
    config_class = GPTConfig
    base_model_prefix = "gpt"

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


class GPTModel(GPTPretrainedModel):
    def __init__(self, config: GPTConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings, config.hidden_size)
        self.layers = nn.ModuleList(
            [GPTDecoderLayer(config) for _ in range(config.num_hidden_layers)]
        )
        self.final_norm = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        B, S = input_ids.shape
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        pos = torch.arange(offset, offset + S, device=input_ids.device)
        x = self.embed_tokens(input_ids) + self.position_embeddings(pos)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.final_norm(x)
        if use_cache:
            return x, presents
        return x


class GPTPretrainingCriterion(nn.Module):
    def __init__(self, config: GPTConfig):
        super().__init__()
        self.ignore_index = -100

    def forward(self, logits, labels):
        return ops.cross_entropy(
            logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
            self.ignore_index, reduction="mean",
        )


class GPTForCausalLM(GPTPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: GPTConfig):
        super().__init__(config)
        self.gpt = GPTModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        self.criterion = GPTPretrainingCriterion(config)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_output_embeddings(self):
        return self.lm_head

    def get_input_embeddings(self):
        return self.gpt.embed_tokens

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, position_ids=None, **kwargs):
        out = self.gpt(input_ids, past_key_values, use_cache)
        if use_cache:
            hidden, presents = out
        else:
            hidden, presents = out, None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = self.criterion(logits, labels)
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
