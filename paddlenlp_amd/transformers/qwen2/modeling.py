"""Qwen2 model family (reference: paddlenlp/transformers/qwen2/modeling.py).

Llama architecture with bias on the q/k/v projections and optional tied
embeddings.  Reuses the Llama decoder stack; only the attention projections
differ, handled by a config-driven subclass.
"""
from __future__ import annotations

import torch.nn as nn

from ...generation import GenerationConfig, GenerationMixin
from ..llama.modeling import (
    LlamaAttention,
    LlamaDecoderLayer,
    LlamaModel,
    LlamaPretrainedModel,
    LlamaPretrainingCriterion,
    _Linear,
    _linear_classes,
)
from .configuration import Qwen2Config

__all__ = ["Qwen2Model", "Qwen2ForCausalLM"]


class Qwen2Attention(LlamaAttention):
    def __init__(self, config: Qwen2Config, layer_idx: int = 0):
        super().__init__(config, layer_idx)
        if getattr(config, "attention_bias", True):
            # re-create the projections with bias (llama builds them without)
            Column, Row = _linear_classes(config)
            d = self.head_dim
            q_out = self.num_heads * d
            kv_out = self.num_kv_heads * d
            if config.fuse_attention_qkv:
                self.qkv_proj = Column(self.hidden_size, q_out + 2 * kv_out, bias=True)
            else:
                self.q_proj = Column(self.hidden_size, q_out, bias=True)
                self.k_proj = Column(self.hidden_size, kv_out, bias=True)
                self.v_proj = Column(self.hidden_size, kv_out, bias=True)


class Qwen2DecoderLayer(LlamaDecoderLayer):
    def __init__(self, config: Qwen2Config, layer_idx: int = 0):
        super().__init__(config, layer_idx)
        self.self_attn = Qwen2Attention(config, layer_idx)


class Qwen2PretrainedModel(LlamaPretrainedModel):
    config_class = Qwen2Config
    base_model_prefix = "qwen2"


class Qwen2Model(Qwen2PretrainedModel, LlamaModel):
    def __init__(self, config: Qwen2Config):
        LlamaModel.__init__(self, config)
        self.layers = nn.ModuleList(
            [Qwen2DecoderLayer(config, i) for i in range(config.num_hidden_layers)]
        )


class Qwen2ForCausalLM(Qwen2PretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: Qwen2Config):
        super().__init__(config)
        self.qwen2 = Qwen2Model(config)
        self.lm_head = _Linear(config.hidden_size, config.vocab_size, bias=False)
        self.criterion = LlamaPretrainingCriterion(config)
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.qwen2.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.qwen2(input_ids=input_ids, past_key_values=past_key_values,
                         use_cache=use_cache)
        if use_cache:
            hidden, presents = out
        else:
            hidden, presents = out, None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = self.criterion(logits, labels)
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)
