"""ChatGLM2 model family (reference: paddlenlp/transformers/chatglm_v2/modeling.py).

GLM-2 decoder: RMSNorm, GQA ("multi-query group") attention with biased QKV
and rotary position embedding applied to HALF the head dims in interleaved
(even, odd) pairs, SwiGLU FFN, tied-norm pre-LN layout.  Attention scores
run through the flash-attention seam ([B, S, H, D] layout) — only the rope
application is family-specific.
"""
from __future__ import annotations

import torch
import torch.nn as nn

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..llama.modeling import LlamaRMSNorm
from ..model_utils import PretrainedModel
from .configuration import ChatGLMv2Config

__all__ = ["ChatGLMv2Model", "ChatGLMv2ForCausalLM"]


def _glm_rope(x: torch.Tensor, cos: torch.Tensor, sin: torch.Tensor):
    """Interleaved-pair rotary on the FIRST half of the head dim.

    x: [B, S, H, D]; cos/sin: [S, D//4] (the rotary half has D//2 dims =
    D//4 pairs)."""
    D = x.shape[-1]
    rot, passthrough = x[..., :D // 2], x[..., D // 2:]
    x0 = rot[..., 0::2]
    x1 = rot[..., 1::2]
    c = cos[None, :, None, :]
    s = sin[None, :, None, :]
    out0 = x0 * c - x1 * s
    out1 = x1 * c + x0 * s
    rotated = torch.stack([out0, out1], dim=-1).flatten(-2)
    return torch.cat([rotated, passthrough], dim=-1)


class ChatGLMv2Attention(nn.Module):
    def __init__(self, config: ChatGLMv2Config):
        super().__init__()
        c = config
        self.num_heads = c.num_attention_heads
        self.num_kv_heads = c.multi_query_group_num
        self.head_dim = c.kv_channels
        q_out = self.num_heads * self.head_dim
        kv_out = self.num_kv_heads * self.head_dim
        self.query_key_value = nn.Linear(
            c.hidden_size, q_out + 2 * kv_out, bias=c.add_qkv_bias)
        self.dense = nn.Linear(q_out, c.hidden_size, bias=False)
        self.rope_theta = c.rope_theta
        self.q_out, self.kv_out = q_out, kv_out

    def _cos_sin(self, S, device, dtype, offset):
        # rotary over head_dim/2 dims -> head_dim/4 frequency pairs
        n = self.head_dim // 4
        inv = 1.0 / (self.rope_theta
                     ** (torch.arange(n, device=device).float() / n))
        t = torch.arange(offset, offset + S, device=device).float()
        freqs = torch.outer(t, inv)
        return freqs.cos().to(dtype), freqs.sin().to(dtype)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        B, S, _ = x.shape
        qkv = self.query_key_value(x)
        q, k, v = qkv.split([self.q_out, self.kv_out, self.kv_out], dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim)
        k = k.view(B, S, self.num_kv_heads, self.head_dim)
        v = v.view(B, S, self.num_kv_heads, self.head_dim)
        cos, sin = self._cos_sin(S, x.device, x.dtype, position_offset)
        q = _glm_rope(q, cos, sin)
        k = _glm_rope(k, cos, sin)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v) if use_cache else None
        out = ops.flash_attention(q, k, v, causal=True)
        out = self.dense(out.reshape(B, S, self.q_out))
        if use_cache:
            return out, present
        return out


class ChatGLMv2MLP(nn.Module):
    def __init__(self, config: ChatGLMv2Config):
        super().__init__()
        h, i = config.hidden_size, config.ffn_hidden_size
        self.dense_h_to_4h = nn.Linear(h, 2 * i, bias=False)  # [gate | up]
        self.dense_4h_to_h = nn.Linear(i, h, bias=False)

    def forward(self, x):
        return self.dense_4h_to_h(ops.swiglu(self.dense_h_to_4h(x)))


class ChatGLMv2Block(nn.Module):
    def __init__(self, config: ChatGLMv2Config):
        super().__init__()
        self.input_layernorm = LlamaRMSNorm(config)
        self.self_attention = ChatGLMv2Attention(config)
        self.post_attention_layernorm = LlamaRMSNorm(config)
        self.mlp = ChatGLMv2MLP(config)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        attn = self.self_attention(self.input_layernorm(x), past_key_value,
                                   use_cache, position_offset)
        if use_cache:
            attn, present = attn
        x = x + attn
        x = x + self.mlp(self.post_attention_layernorm(x))
        if use_cache:
            return x, present
        return x


class ChatGLMv2PretrainedModel(PretrainedModel):
    config_class = ChatGLMv2Config
    base_model_prefix = "chatglm_v2"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)


class ChatGLMv2Model(ChatGLMv2PretrainedModel):
    def __init__(self, config: ChatGLMv2Config):
        super().__init__(config)
        self.embedding = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [ChatGLMv2Block(config) for _ in range(config.num_hidden_layers)])
        self.final_layernorm = LlamaRMSNorm(config)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embedding

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embedding(input_ids)
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
        x = self.final_layernorm(x)
        if use_cache:
            return x, presents
        return x


class ChatGLMv2ForCausalLM(ChatGLMv2PretrainedModel, GenerationMixin):
    def __init__(self, config: ChatGLMv2Config):
        super().__init__(config)
        self.chatglm_v2 = ChatGLMv2Model(config)
        self.output_layer = nn.Linear(config.hidden_size, config.vocab_size,
                                      bias=False)
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.chatglm_v2.embedding

    def get_output_embeddings(self):
        return self.output_layer

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.chatglm_v2(input_ids, past_key_values, use_cache)
        if use_cache:
            hidden, presents = out
        else:
            hidden, presents = out, None
        logits = self.output_layer(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                -100, reduction="mean")
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
