"""ERNIE-GEN (reference: paddlenlp/transformers/ernie_gen/modeling.py).

Generation pre-training on an ERNIE-style encoder driven by an
ATTENTION-BIAS matrix instead of fixed masking (reference attn forward
:87-120): the caller supplies src/tgt spans, and the model builds the
span-infilling bias — source tokens attend bidirectionally among
themselves, target tokens attend to the source plus causally to earlier
target tokens.  Decoding feeds the growing target through the same
bias.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import EncoderLayer, LMPredictionHead, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ErnieGenConfig", "ErnieGenModel", "ErnieGenForGeneration"]


class ErnieGenConfig(PretrainedConfig):
    model_type = "ernie_gen"

    def __init__(self, vocab_size=18000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=513,
                 type_vocab_size=4, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class _BiasLayer(EncoderLayer):
    """EncoderLayer whose attention takes a full additive bias matrix."""

    def forward(self, x, attn_bias=None):
        B, S, H = x.shape
        attn = self.self_attn
        q, k, v = attn.qkv_proj(x).chunk(3, dim=-1)
        shp = (B, S, attn.num_heads, attn.head_dim)
        o = F.scaled_dot_product_attention(
            q.view(shp).transpose(1, 2), k.view(shp).transpose(1, 2),
            v.view(shp).transpose(1, 2),
            attn_mask=attn_bias)
        x = self.attn_norm(x + attn.out_proj(o.transpose(1, 2).reshape(B, S, H)))
        return self.mlp_norm(x + self.fc_out(self.act(self.fc_in(x))))


def build_infill_bias(src_len, tgt_len, device, dtype):
    """[1,1,S,S] bias: src bidirectional; tgt sees src + causal tgt."""
    S = src_len + tgt_len
    i = torch.arange(S, device=device).view(S, 1)
    j = torch.arange(S, device=device).view(1, S)
    src_q = i < src_len
    vis = torch.where(src_q, j < src_len,                 # src -> src only
                      (j < src_len) | (j <= i))           # tgt -> src+causal
    return torch.where(vis, torch.zeros((), device=device, dtype=dtype),
                       torch.full((), torch.finfo(dtype).min,
                                  device=device, dtype=dtype))[None, None]


class ErnieGenPretrainedModel(PretrainedModel):
    config_class = ErnieGenConfig
    base_model_prefix = "ernie_gen"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieGenModel(ErnieGenPretrainedModel):
    def __init__(self, config: ErnieGenConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [_BiasLayer(config) for _ in range(config.num_hidden_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, token_type_ids=None, attn_bias=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        for layer in self.layers:
            x = layer(x, attn_bias)
        return x


class ErnieGenForGeneration(ErnieGenPretrainedModel):
    _tied_weights_keys = ["mlm.decoder.weight"]

    def __init__(self, config: ErnieGenConfig):
        super().__init__(config)
        self.ernie_gen = ErnieGenModel(config)
        self.mlm = LMPredictionHead(config, self.ernie_gen.embeddings.weight)

    def forward(self, input_ids, src_len, token_type_ids=None, labels=None):
        """input_ids = [source ; target]; src_len marks the boundary."""
        S = input_ids.shape[1]
        bias = build_infill_bias(src_len, S - src_len, input_ids.device,
                                 self.mlm.decoder.weight.dtype
                                 if hasattr(self.mlm, "decoder")
                                 else torch.float32)
        seq = self.ernie_gen(input_ids, token_type_ids, bias)
        logits = self.mlm(seq[:, src_len:])
        if labels is not None:
            # labels align with the target region, next-token shifted
            loss = F.cross_entropy(
                logits[:, :-1].reshape(-1, self.config.vocab_size),
                labels[:, 1:].reshape(-1), ignore_index=-100)
            return loss, logits
        return logits
