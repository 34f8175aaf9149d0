"""GLM (reference: paddlenlp/transformers/glm/modeling.py).

Autoregressive blank infilling: 2-D positions (absolute + in-block,
reference GLMStack :294-379), a per-sample ltor split point — tokens
before the split attend bidirectionally, tokens at/after it attend
causally (the GLM [sMASK]/[gMASK] generation scheme) — and a tied LM
head for conditional generation over the filled blanks.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...generation import GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["GLMConfig", "GLMModel", "GLMForConditionalGeneration"]


class GLMConfig(PretrainedConfig):
    model_type = "glm"

    def __init__(self, vocab_size=50304, hidden_size=1024,
                 num_hidden_layers=24, num_attention_heads=16,
                 max_position_embeddings=512, block_position_encoding=True,
                 layernorm_epsilon=1e-5, hidden_dropout_prob=0.1,
                 initializer_range=0.02, pad_token_id=50000, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.max_position_embeddings = max_position_embeddings
        self.block_position_encoding = block_position_encoding
        self.layernorm_epsilon = layernorm_epsilon
        self.hidden_dropout_prob = hidden_dropout_prob
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class GLMBlock(nn.Module):
    """Pre-norm block (reference GLMBlock :213)."""

    def __init__(self, config: GLMConfig):
        super().__init__()
        h = config.hidden_size
        self.input_layernorm = nn.LayerNorm(h, eps=config.layernorm_epsilon)
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.query_key_value = nn.Linear(h, 3 * h)
        self.dense = nn.Linear(h, h)
        self.post_attention_layernorm = nn.LayerNorm(
            h, eps=config.layernorm_epsilon)
        self.mlp_in = nn.Linear(h, 4 * h)
        self.mlp_out = nn.Linear(4 * h, h)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, ltor_mask, past_key_value=None, use_cache=False):
        B, S, H = x.shape
        h = self.input_layernorm(x)
        q, k, v = self.query_key_value(h).chunk(3, dim=-1)
        shp = (B, S, self.num_heads, self.head_dim)
        q = q.view(shp).transpose(1, 2)
        k = k.view(shp).transpose(1, 2)
        v = v.view(shp).transpose(1, 2)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=2)
            v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None
        attn = F.scaled_dot_product_attention(q, k, v,
                                              attn_mask=ltor_mask)
        attn = self.dense(attn.transpose(1, 2).reshape(B, S, H))
        x = x + self.dropout(attn)
        h = self.post_attention_layernorm(x)
        y = self.mlp_out(F.gelu(self.mlp_in(h)))
        x = x + self.dropout(y)
        if use_cache:
            return x, present
        return x


class GLMPretrainedModel(PretrainedModel):
    config_class = GLMConfig
    base_model_prefix = "glm"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class GLMModel(GLMPretrainedModel):
    def __init__(self, config: GLMConfig):
        super().__init__(config)
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        if config.block_position_encoding:
            # block positions get their OWN table (reference :296-302)
            self.block_position_embeddings = nn.Embedding(
                config.max_position_embeddings + 1, h)
        self.layers = nn.ModuleList(
            [GLMBlock(config) for _ in range(config.num_hidden_layers)])
        self.final_layernorm = nn.LayerNorm(h, eps=config.layernorm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def _build_mask(self, B, S, split, device, dtype, past_len=0):
        """[B,1,S,klen] additive mask: key j visible to query i iff
        j < split[b] (context, bidirectional) or j <= i (causal)."""
        i = torch.arange(past_len, past_len + S, device=device).view(1, S, 1)
        j = torch.arange(past_len + S, device=device).view(1, 1, -1)
        sp = split.view(B, 1, 1)
        visible = (j < sp) | (j <= i)
        return torch.where(visible, torch.zeros(1, device=device, dtype=dtype),
                           torch.full((1,), torch.finfo(dtype).min,
                                      device=device, dtype=dtype)).unsqueeze(1)

    def forward(self, input_ids, position_ids=None, attention_mask=None,
                past_key_values=None, use_cache=False):
        B, S = input_ids.shape
        past_len = (past_key_values[0][0].shape[2]
                    if past_key_values is not None else 0)
        x = self.word_embeddings(input_ids)
        if position_ids is None:
            pos = torch.arange(past_len, past_len + S,
                               device=input_ids.device)
            position_ids = torch.stack(
                [pos, torch.zeros_like(pos)]).unsqueeze(0).expand(B, 2, S)
        if self.config.block_position_encoding:
            # [B, 2, S]: absolute + in-block positions (reference :371-379)
            abs_pos, blk_pos = position_ids[:, 0], position_ids[:, 1]
            x = x + self.position_embeddings(abs_pos) \
                  + self.block_position_embeddings(blk_pos)
        else:
            x = x + self.position_embeddings(position_ids)
        # attention_mask: [B] split index (reference passes the sep
        # position as the "ltor" boundary); None = fully causal
        if attention_mask is None:
            split = torch.zeros(B, dtype=torch.long, device=x.device)
        elif attention_mask.dim() == 1:
            split = attention_mask.long()
        else:
            split = attention_mask.long().view(B, -1)[:, 0]
        mask = self._build_mask(B, S, split, x.device, x.dtype, past_len)
        presents = [] if use_cache else None
        for li, layer in enumerate(self.layers):
            past = past_key_values[li] if past_key_values is not None else None
            out = layer(x, mask, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.final_layernorm(x)
        if use_cache:
            return x, presents
        return x


class GLMForConditionalGeneration(GLMPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: GLMConfig):
        super().__init__(config)
        self.glm = GLMModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=False)
        self.lm_head.weight = self.glm.word_embeddings.weight

    def forward(self, input_ids, position_ids=None, attention_mask=None,
                labels=None, past_key_values=None, use_cache=False, **kwargs):
        out = self.glm(input_ids, position_ids, attention_mask,
                       past_key_values, use_cache)
        presents = None
        if use_cache:
            out, presents = out
        logits = self.lm_head(out)
        if labels is not None:
            # labels are pre-shifted by the caller (framework convention)
            loss = F.cross_entropy(
                logits.reshape(-1, self.config.vocab_size),
                labels.reshape(-1), ignore_index=-100)
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
