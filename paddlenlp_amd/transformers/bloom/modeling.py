"""BLOOM model family (reference: paddlenlp/transformers/bloom/modeling.py).

GPT-style pre-LN decoder with ALiBi position bias instead of positional
embeddings (slopes shared with long_sequence_strategies), an embedding
LayerNorm, GELU MLP (4x), tied LM head.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..long_sequence_strategies import AttentionWithLinearBias
from ..model_utils import PretrainedModel
from .configuration import BloomConfig

__all__ = ["BloomModel", "BloomForCausalLM"]


class BloomAttention(nn.Module):
    def __init__(self, config: BloomConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.query_key_value = nn.Linear(h, 3 * h, bias=True)
        self.dense = nn.Linear(h, h, bias=True)

    def forward(self, x, alibi, past_key_value=None, use_cache=False):
        B, S, H = x.shape
        q, k, v = self.query_key_value(x).chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        q = q.view(shape).transpose(1, 2)
        k = k.view(shape).transpose(1, 2)
        v = v.view(shape).transpose(1, 2)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=2)
            v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None
        Skv = k.shape[2]
        # alibi: [1, heads, 1, Skv] + causal mask
        mask = alibi[:, :, :, :Skv].to(x.dtype)
        if S > 1:
            cmask = torch.full((S, Skv), float("-inf"), device=x.device,
                               dtype=x.dtype).triu(Skv - S + 1)
            mask = mask + cmask
        out = F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
        out = self.dense(out.transpose(1, 2).reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class BloomBlock(nn.Module):
    def __init__(self, config: BloomConfig):
        super().__init__()
        h = config.hidden_size
        self.input_layernorm = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.self_attention = BloomAttention(config)
        self.post_attention_layernorm = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.dense_h_to_4h = nn.Linear(h, 4 * h)
        self.dense_4h_to_h = nn.Linear(4 * h, h)
        self.post_residual_ln = config.apply_residual_connection_post_layernorm

    def forward(self, x, alibi, past_key_value=None, use_cache=False):
        ln = self.input_layernorm(x)
        residual = ln if self.post_residual_ln else x
        attn = self.self_attention(ln, alibi, past_key_value, use_cache)
        if use_cache:
            attn, present = attn
        x = residual + attn
        ln = self.post_attention_layernorm(x)
        residual = ln if self.post_residual_ln else x
        x = residual + self.dense_4h_to_h(
            F.gelu(self.dense_h_to_4h(ln), approximate="tanh"))
        if use_cache:
            return x, present
        return x


class BloomPretrainedModel(PretrainedModel):
    config_class = BloomConfig
    base_model_prefix = "bloom"

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


class BloomModel(BloomPretrainedModel):
    def __init__(self, config: BloomConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size, config.hidden_size)
        self.word_embeddings_layernorm = nn.LayerNorm(
            config.hidden_size, eps=config.layer_norm_epsilon)
        self.h = nn.ModuleList(
            [BloomBlock(config) for _ in range(config.num_hidden_layers)])
        self.ln_f = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def _alibi(self, total_len: int, device):
        # AttentionWithLinearBias.bias -> [H, 1, S]; add the batch dim
        bias = AttentionWithLinearBias.bias(
            self.config.num_attention_heads, total_len, device)
        return bias.unsqueeze(0)  # [1, H, 1, S] additive

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None:
            past_len = past_key_values[0][0].shape[2]
        x = self.word_embeddings_layernorm(self.word_embeddings(input_ids))
        alibi = self._alibi(past_len + input_ids.shape[1], input_ids.device)
        presents = [] if use_cache else None
        for i, block in enumerate(self.h):
            past = past_key_values[i] if past_key_values is not None else None
            out = block(x, alibi, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.ln_f(x)
        if use_cache:
            return x, presents
        return x


class BloomForCausalLM(BloomPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: BloomConfig):
        super().__init__(config)
        self.bloom = BloomModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.bloom.word_embeddings.weight
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.bloom.word_embeddings.weight

    def get_input_embeddings(self):
        return self.bloom.word_embeddings

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.bloom(input_ids, past_key_values, use_cache)
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
