"""OPT model family (reference: paddlenlp/transformers/opt/modeling.py).

GPT-2 architecture with OPT specifics: learned positions offset by +2
reserved rows, ReLU MLP, pre-LN (do_layer_norm_before) with a final
LayerNorm, tied LM head.  Attention reuses the GPT fused-QKV path over the
flash-attention seam.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..gpt.modeling import GPTAttention
from ..model_utils import PretrainedModel
from .configuration import OPTConfig

__all__ = ["OPTModel", "OPTForCausalLM"]


class OPTLearnedPositionalEmbedding(nn.Embedding):
    OFFSET = 2

    def __init__(self, num_positions: int, d_model: int):
        super().__init__(num_positions + self.OFFSET, d_model)

    def forward(self, seq_len: int, past_len: int = 0):
        pos = torch.arange(past_len, past_len + seq_len,
                           device=self.weight.device)
        return super().forward(pos + self.OFFSET)


class OPTDecoderLayer(nn.Module):
    def __init__(self, config: OPTConfig):
        super().__init__()
        h = config.hidden_size
        self.pre_ln = config.do_layer_norm_before
        self.self_attn = GPTAttention(config)
        self.self_attn_layer_norm = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.fc1 = nn.Linear(h, config.intermediate_size)
        self.fc2 = nn.Linear(config.intermediate_size, h)
        self.final_layer_norm = nn.LayerNorm(h, eps=config.layer_norm_epsilon)

    def forward(self, x, past_key_value=None, use_cache=False):
        residual = x
        h = self.self_attn_layer_norm(x) if self.pre_ln else x
        attn = self.self_attn(h, past_key_value, use_cache)
        if use_cache:
            attn, present = attn
        x = residual + attn
        if not self.pre_ln:
            x = self.self_attn_layer_norm(x)
        residual = x
        h = self.final_layer_norm(x) if self.pre_ln else x
        x = residual + self.fc2(F.relu(self.fc1(h)))
        if not self.pre_ln:
            x = self.final_layer_norm(x)
        if use_cache:
            return x, present
        return x


class OPTPretrainedModel(PretrainedModel):
    config_class = OPTConfig
    base_model_prefix = "opt"

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


class OPTModel(OPTPretrainedModel):
    def __init__(self, config: OPTConfig):
        super().__init__(config)
        self.embed_tokens = nn.Embedding(config.vocab_size, config.hidden_size)
        self.embed_positions = OPTLearnedPositionalEmbedding(
            config.max_position_embeddings, config.hidden_size)
        self.layers = nn.ModuleList(
            [OPTDecoderLayer(config) for _ in range(config.num_hidden_layers)])
        self.final_layer_norm = (
            nn.LayerNorm(config.hidden_size, eps=config.layer_norm_epsilon)
            if config.do_layer_norm_before else None)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embed_tokens

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None:
            past_len = past_key_values[0][0].shape[1]
        x = self.embed_tokens(input_ids) + self.embed_positions(
            input_ids.shape[1], past_len)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        if self.final_layer_norm is not None:
            x = self.final_layer_norm(x)
        if use_cache:
            return x, presents
        return x


class OPTForCausalLM(OPTPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: OPTConfig):
        super().__init__(config)
        self.opt = OPTModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.opt.embed_tokens.weight
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.opt.embed_tokens.weight

    def get_input_embeddings(self):
        return self.opt.embed_tokens

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.opt(input_ids, past_key_values, use_cache)
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
