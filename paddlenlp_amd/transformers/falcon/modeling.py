"""Falcon model family (reference: paddlenlp/transformers/rw/ — the
falcon-7b/40b "RW" architecture).

Parallel-residual decoder: attention and MLP both read the SAME layernormed
input and their outputs sum into one residual (parallel_attn=True, the
falcon-7b form; parallel_attn=False gives the sequential two-LN layout).
MQA/GQA rotary attention reuses the Llama GQA path.
"""
from __future__ import annotations

import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..llama.modeling import LlamaAttention
from ..model_utils import PretrainedModel
from .configuration import FalconConfig

__all__ = ["FalconModel", "FalconForCausalLM"]


class FalconMLP(nn.Module):
    def __init__(self, config: FalconConfig):
        super().__init__()
        h = config.hidden_size
        self.dense_h_to_4h = nn.Linear(h, config.intermediate_size, bias=config.bias)
        self.dense_4h_to_h = nn.Linear(config.intermediate_size, h, bias=config.bias)

    def forward(self, x):
        return self.dense_4h_to_h(F.gelu(self.dense_h_to_4h(x)))


class FalconDecoderLayer(nn.Module):
    def __init__(self, config: FalconConfig, layer_idx: int = 0):
        super().__init__()
        h = config.hidden_size
        self.parallel_attn = config.parallel_attn
        self.self_attention = LlamaAttention(config, layer_idx)
        self.mlp = FalconMLP(config)
        self.input_layernorm = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        if not config.parallel_attn:
            self.post_attention_layernorm = nn.LayerNorm(
                h, eps=config.layer_norm_epsilon)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        ln = self.input_layernorm(x)
        attn = self.self_attention(ln, None, None, past_key_value, use_cache,
                                   position_offset)
        if use_cache:
            attn, present = attn
        if self.parallel_attn:
            # one residual: x + attn(ln(x)) + mlp(ln(x))
            x = x + attn + self.mlp(ln)
        else:
            x = x + attn
            x = x + self.mlp(self.post_attention_layernorm(x))
        if use_cache:
            return x, present
        return x


class FalconPretrainedModel(PretrainedModel):
    config_class = FalconConfig
    base_model_prefix = "falcon"

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


class FalconModel(FalconPretrainedModel):
    def __init__(self, config: FalconConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size, config.hidden_size)
        self.h = nn.ModuleList(
            [FalconDecoderLayer(config, i)
             for i in range(config.num_hidden_layers)])
        self.ln_f = nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.word_embeddings(input_ids)
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        presents = [] if use_cache else None
        for i, layer in enumerate(self.h):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, past, use_cache, offset)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.ln_f(x)
        if use_cache:
            return x, presents
        return x


class FalconForCausalLM(FalconPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: FalconConfig):
        super().__init__(config)
        self.falcon = FalconModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.falcon.word_embeddings.weight
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.falcon.word_embeddings.weight

    def get_input_embeddings(self):
        return self.falcon.word_embeddings

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.falcon(input_ids, past_key_values, use_cache)
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
