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

from ... import ops
from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
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
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        self.qkv = nn.Linear(h, 3 * h)
        self.proj = nn.Linear(h, h)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.dropout = nn.Dropout(config.resid_pdrop)

    def forward(self, x):
        B, S, H = x.shape
        q, k, v = self.qkv(self.ln1(x)).chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        out = ops.flash_attention(q.view(shape), k.view(shape), v.view(shape),
                                  causal=True)
        x = x + self.dropout(self.proj(out.reshape(B, S, H)))
        y = self.fc_out(F.relu(self.fc_in(self.ln2(x))))
        return x + self.dropout(y)


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

    def forward(self, input_ids, attention_mask=None):
        B, S = input_ids.shape
        # reference scales token embeddings by sqrt(d) before adding PE
        x = self.w(input_ids) * math.sqrt(self.config.hidden_size)
        x = x + self._sinusoid(S, x.device, x.dtype)
        x = self.dropout(x)
        for block in self.h:
            x = block(x)
        return self.layernorm(x)


class CTRLLMHeadModel(CTRLPretrainedModel):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: CTRLConfig):
        super().__init__(config)
        self.transformer = CTRLModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=True)
        self.lm_head.weight = self.transformer.w.weight

    def forward(self, input_ids, attention_mask=None, labels=None):
        logits = self.lm_head(self.transformer(input_ids, attention_mask))
        if labels is not None:
            loss = F.cross_entropy(
                logits[:, :-1].reshape(-1, self.config.vocab_size),
                labels[:, 1:].reshape(-1), ignore_index=-100)
            return loss, logits
        return logits

    def prepare_inputs_for_generation(self, input_ids, **kwargs):
        return {"input_ids": input_ids}


class CTRLForSequenceClassification(CTRLPretrainedModel):
    def __init__(self, config: CTRLConfig):
        super().__init__(config)
        self.transformer = CTRLModel(config)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels,
                                    bias=False)

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
