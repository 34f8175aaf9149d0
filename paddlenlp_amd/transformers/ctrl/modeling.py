"""CTRL (reference: paddlenlp/transformers/ctrl/modeling.py).

Conditional causal LM: GPT-shaped decoder whose vocabulary begins with
control codes; pre-norm layers, sinusoidal positions scaled by
sqrt(d_model) on the embeddings, tied LM head WITH bias (the reference's
`lm_head` keeps a bias unlike GPT-2).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...generation import GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..gpt.modeling import GPTAttention
from ..model_utils import PretrainedModel

__all__ = ["CTRLConfig", "CTRLModel", "CTRLLMHeadModel",
           "CTRLForSequenceClassification"]


class CTRLConfig(PretrainedConfig):
    model_type = "ctrl"

    attribute_map = {"n_embd": "hidden_size", "n_layer": "num_hidden_layers",
                     "n_head": "num_attention_heads",
                     "num_classes": "num_labels"}

    def __init__(self, vocab_size=246534, hidden_size=1280,
                 num_hidden_layers=48, num_attention_heads=16,
                 intermediate_size=8192, max_position_embeddings=256,
                 resid_pdrop=0.1, embd_pdrop=0.1,
                 layer_norm_epsilon=1e-6, initializer_range=0.02,
                 pad_token_id=None, num_labels=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.max_position_embeddings = max_position_embeddings
        self.resid_pdrop = resid_pdrop
        self.embd_pdrop = embd_pdrop
        self.layer_norm_epsilon = layer_norm_epsilon
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class CTRLBlock(nn.Module):
    def __init__(self, config: CTRLConfig):
        super().__init__()
        h = config.hidden_size
        self.ln1 = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.ln2 = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.attn = GPTAttention(config)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.dropout = nn.Dropout(config.resid_pdrop)

    def forward(self, x, past_key_value=None, use_cache=False):
        out = self.attn(self.ln1(x), past_key_value, use_cache)
        present = None
        if use_cache:
            out, present = out
        x = x + self.dropout(out)
        y = self.fc_out(F.relu(self.fc_in(self.ln2(x))))
        x = x + self.dropout(y)
        if use_cache:
            return x, present
        return x


class CTRLPretrainedModel(PretrainedModel):
    config_class = CTRLConfig
    base_model_prefix = "ctrl"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class CTRLModel(CTRLPretrainedModel):
    def __init__(self, config: CTRLConfig):
        super().__init__(config)
        self.w = nn.Embedding(config.vocab_size, config.hidden_size)
        self.dropout = nn.Dropout(config.embd_pdrop)
        self.h = nn.ModuleList(
            [CTRLBlock(config) for _ in range(config.num_hidden_layers)])
        self.layernorm = nn.LayerNorm(config.hidden_size,
                                      eps=config.layer_norm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.w

    def _sinusoid(self, S, device, dtype):
        h = self.config.hidden_size
        pos = torch.arange(S, device=device).float()
        inv = 1.0 / (10000 ** (torch.arange(0, h, 2, device=device).float() / h))
        ang = torch.outer(pos, inv)
        pe = torch.zeros(S, h, device=device)
        pe[:, 0::2] = ang.sin()
        pe[:, 1::2] = ang.cos()
        return pe.to(dtype)

    def forward(self, input_ids, attention_mask=None, past_key_values=None,
                use_cache=False):
        B, S = input_ids.shape
        past_len = (past_key_values[0][0].shape[1]
                    if past_key_values is not None else 0)
        # reference scales token embeddings by sqrt(d) before adding PE
        x = self.w(input_ids) * math.sqrt(self.config.hidden_size)
        pe = self._sinusoid(past_len + S, x.device, x.dtype)[past_len:]
        x = self.dropout(x + pe)
        presents = [] if use_cache else None
        for i, block in enumerate(self.h):
            past = past_key_values[i] if past_key_values is not None else None
            out = block(x, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.layernorm(x)
        if use_cache:
            return x, presents
        return x


class CTRLLMHeadModel(CTRLPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: CTRLConfig):
        super().__init__(config)
        self.transformer = CTRLModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=True)
        self.init_weights()
        self.lm_head.weight = self.transformer.w.weight

    def forward(self, input_ids, attention_mask=None, labels=None,
                past_key_values=None, use_cache=False, **kwargs):
        out = self.transformer(input_ids, attention_mask,
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


class CTRLForSequenceClassification(CTRLPretrainedModel):
    def __init__(self, config: CTRLConfig):
        super().__init__(config)
        self.transformer = CTRLModel(config)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels,
                                    bias=False)
        self.init_weights()

    def forward(self, input_ids, attention_mask=None, labels=None):
        seq = self.transformer(input_ids, attention_mask)
        # classify on the last non-pad token (reference behavior)
        if self.config.pad_token_id is not None:
            last = (input_ids != self.config.pad_token_id).long().sum(1) - 1
        elif attention_mask is not None:
            last = attention_mask.long().sum(1) - 1
        else:
            last = torch.full((input_ids.shape[0],), input_ids.shape[1] - 1,
                              dtype=torch.long, device=input_ids.device)
        pooled = seq[torch.arange(seq.shape[0], device=seq.device), last]
        logits = self.classifier(pooled)
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
