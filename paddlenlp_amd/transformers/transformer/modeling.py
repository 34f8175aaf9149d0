"""Classic Transformer NMT model (reference:
paddlenlp/transformers/transformer/modeling.py).

The original encoder-decoder: sinusoidal positions, scaled embeddings
shared between encoder/decoder (+ tied output projection when
`weight_sharing`), label smoothing in the training loss, greedy/beam
decode through the framework GenerationMixin.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...generation import GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["TransformerConfig", "TransformerModel"]


class TransformerConfig(PretrainedConfig):
    model_type = "transformer"

    def __init__(self, src_vocab_size=30000, trg_vocab_size=30000,
                 hidden_size=512, num_encoder_layers=6,
                 num_decoder_layers=6, num_attention_heads=8,
                 intermediate_size=2048, dropout=0.1,
                 max_position_embeddings=256, weight_sharing=True,
                 label_smooth_eps=0.1, bos_token_id=0, eos_token_id=1,
                 pad_token_id=0, initializer_range=0.02, **kwargs):
        super().__init__(**kwargs)
        self.src_vocab_size = src_vocab_size
        self.trg_vocab_size = trg_vocab_size
        self.vocab_size = trg_vocab_size
        self.hidden_size = hidden_size
        self.num_encoder_layers = num_encoder_layers
        self.num_decoder_layers = num_decoder_layers
        self.num_hidden_layers = num_encoder_layers + num_decoder_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.dropout = dropout
        self.max_position_embeddings = max_position_embeddings
        self.weight_sharing = weight_sharing
        self.label_smooth_eps = label_smooth_eps
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.pad_token_id = pad_token_id
        self.initializer_range = initializer_range


def _sinusoid(S, h, device, dtype):
    pos = torch.arange(S, device=device).float()
    inv = 1.0 / (10000 ** (torch.arange(0, h, 2, device=device).float() / h))
    ang = torch.outer(pos, inv)
    pe = torch.zeros(S, h, device=device)
    pe[:, 0::2] = ang.sin()
    pe[:, 1::2] = ang.cos()
    return pe.to(dtype)


class TransformerPretrainedModel(PretrainedModel):
    config_class = TransformerConfig
    base_model_prefix = "transformer"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class TransformerModel(TransformerPretrainedModel, GenerationMixin):
    def __init__(self, config: TransformerConfig):
        super().__init__(config)
        h = config.hidden_size
        self.src_embedding = nn.Embedding(config.src_vocab_size, h,
                                          padding_idx=config.pad_token_id)
        if config.weight_sharing:
            assert config.src_vocab_size == config.trg_vocab_size
            self.trg_embedding = self.src_embedding
        else:
            self.trg_embedding = nn.Embedding(
                config.trg_vocab_size, h, padding_idx=config.pad_token_id)
        # torch-native encoder/decoder stacks (the classic architecture
        # maps 1:1 onto nn.Transformer's post-norm layers)
        self.transformer = nn.Transformer(
            d_model=h, nhead=config.num_attention_heads,
            num_encoder_layers=config.num_encoder_layers,
            num_decoder_layers=config.num_decoder_layers,
            dim_feedforward=config.intermediate_size,
            dropout=config.dropout, batch_first=True)
        self.project_out = nn.Linear(h, config.trg_vocab_size, bias=False)
        self.init_weights()
        if config.weight_sharing:
            self.project_out.weight = self.trg_embedding.weight

    def get_input_embeddings(self):
        return self.src_embedding

    def _embed(self, table, ids):
        h = self.config.hidden_size
        x = table(ids) * math.sqrt(h)
        return x + _sinusoid(ids.shape[1], h, ids.device, x.dtype)

    def forward(self, input_ids, decoder_input_ids=None, labels=None,
                attention_mask=None, **kwargs):
        if decoder_input_ids is None and labels is not None:
            bos = torch.full((labels.shape[0], 1), self.config.bos_token_id,
                             dtype=labels.dtype, device=labels.device)
            decoder_input_ids = torch.cat([bos, labels[:, :-1].clamp(min=0)],
                                          dim=1)
        src = self._embed(self.src_embedding, input_ids)
        tgt = self._embed(self.trg_embedding, decoder_input_ids)
        S = decoder_input_ids.shape[1]
        causal = nn.Transformer.generate_square_subsequent_mask(
            S, device=src.device, dtype=src.dtype)
        pad_mask = None
        if attention_mask is not None:
            pad_mask = attention_mask == 0
        out = self.transformer(src, tgt, tgt_mask=causal,
                               src_key_padding_mask=pad_mask,
                               memory_key_padding_mask=pad_mask)
        logits = self.project_out(out)
        if labels is not None:
            eps = self.config.label_smooth_eps
            loss = F.cross_entropy(
                logits.reshape(-1, self.config.trg_vocab_size),
                labels.reshape(-1), ignore_index=-100,
                label_smoothing=eps)
            return loss, logits
        return logits
