"""ProphetNet (reference: paddlenlp/transformers/prophetnet/modeling.py).

Seq2seq with FUTURE N-GRAM PREDICTION: the decoder runs 1+ngram streams
— the main (causal) stream plus `ngram` predicting streams whose
queries are the main hidden states shifted by a learned ngram embedding
(reference ProphetNetNgramSelfAttention :366-530).  Predict stream g at
position t attends to main-stream keys ≤ t plus its own position, and
is trained on future tokens: the main stream carries the next-token
CE and predict stream g adds an eps-weighted CE on token t+g+1 (the
future-n-gram objective).
"""
from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F

from ...generation import GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, expand_padding_mask, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ProphetNetConfig", "ProphetNetModel",
           "ProphetNetForConditionalGeneration"]


class ProphetNetConfig(PretrainedConfig):
    model_type = "prophetnet"

    def __init__(self, vocab_size=30522, hidden_size=1024,
                 num_encoder_layers=12, num_decoder_layers=12,
                 num_attention_heads=16, intermediate_size=4096,
                 ngram=2, eps=0.1, activation_function="gelu",
                 dropout=0.1, max_position_embeddings=512,
                 initializer_range=0.02, layer_norm_eps=1e-12,
                 pad_token_id=0, bos_token_id=102, eos_token_id=102,
                 decoder_start_token_id=102, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_encoder_layers = num_encoder_layers
        self.num_decoder_layers = num_decoder_layers
        self.num_hidden_layers = num_encoder_layers + num_decoder_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.ngram = ngram
        self.eps = eps
        self.activation_function = activation_function
        self.dropout = dropout
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.decoder_start_token_id = decoder_start_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class _Attention(nn.Module):
    def __init__(self, config, cross=False):
        super().__init__()
        h = config.hidden_size
        self.nh = config.num_attention_heads
        self.dh = config.head_dim
        self.q_proj = nn.Linear(h, h)
        self.k_proj = nn.Linear(h, h)
        self.v_proj = nn.Linear(h, h)
        self.out_proj = nn.Linear(h, h)

    def forward(self, q_in, kv_in, mask=None, causal=False):
        B, Sq, H = q_in.shape
        Sk = kv_in.shape[1]
        q = self.q_proj(q_in).view(B, Sq, self.nh, self.dh).transpose(1, 2)
        k = self.k_proj(kv_in).view(B, Sk, self.nh, self.dh).transpose(1, 2)
        v = self.v_proj(kv_in).view(B, Sk, self.nh, self.dh).transpose(1, 2)
        out = F.scaled_dot_product_attention(q, k, v, attn_mask=mask,
                                             is_causal=causal and mask is None)
        return self.out_proj(out.transpose(1, 2).reshape(B, Sq, H))


class ProphetNetEncoderLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        h = config.hidden_size
        self.attn = _Attention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.ff_in = nn.Linear(h, config.intermediate_size)
        self.ff_out = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.activation_function]

    def forward(self, x, mask=None):
        x = self.attn_norm(x + self.attn(x, x, mask))
        return self.ff_norm(x + self.ff_out(self.act(self.ff_in(x))))


class NgramSelfAttention(nn.Module):
    """Main + predict stream self-attention (reference :366-530).

    Streams are stacked on the sequence axis: [main | pred_1 | ... |
    pred_n], each of length S.  Main is causal over main; predict
    stream g's query at t sees main keys ≤ t and its own (t, g) key."""

    def __init__(self, config):
        super().__init__()
        self.inner = _Attention(config)
        self.ngram = config.ngram

    def forward(self, streams, S):
        n = self.ngram
        B = streams.shape[0]
        total = (1 + n) * S
        # additive mask [total, total]
        i = torch.arange(total, device=streams.device)
        si, ti = i // S, i % S            # stream id, position
        qs, qt = si.view(-1, 1), ti.view(-1, 1)
        ks, kt = si.view(1, -1), ti.view(1, -1)
        vis_main = (ks == 0) & (kt <= qt)              # causal over main
        vis_self = (ks == qs) & (kt == qt) & (qs > 0)  # own-stream key
        visible = torch.where(qs == 0, vis_main, vis_main | vis_self)
        mask = torch.where(
            visible, torch.zeros((), device=streams.device,
                                 dtype=streams.dtype),
            torch.full((), torch.finfo(streams.dtype).min,
                       device=streams.device, dtype=streams.dtype))
        return self.inner(streams, streams, mask[None, None])


class ProphetNetDecoderLayer(nn.Module):
    def __init__(self, config):
        super().__init__()
        h = config.hidden_size
        self.self_attn = NgramSelfAttention(config)
        self.self_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.cross_attn = _Attention(config, cross=True)
        self.cross_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.ff_in = nn.Linear(h, config.intermediate_size)
        self.ff_out = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.activation_function]

    def forward(self, streams, S, enc, enc_mask=None):
        streams = self.self_norm(streams + self.self_attn(streams, S))
        streams = self.cross_norm(
            streams + self.cross_attn(streams, enc, enc_mask))
        return self.ff_norm(
            streams + self.ff_out(self.act(self.ff_in(streams))))


class ProphetNetPretrainedModel(PretrainedModel):
    config_class = ProphetNetConfig
    base_model_prefix = "prophetnet"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ProphetNetModel(ProphetNetPretrainedModel):
    def __init__(self, config: ProphetNetConfig):
        super().__init__(config)
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h,
                                            padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        # one learned embedding per predict stream (reference
        # ngram_embeddings)
        self.ngram_embeddings = nn.Parameter(
            torch.empty(config.ngram, h).normal_(
                std=config.initializer_range))
        self.encoder_layers = nn.ModuleList(
            [ProphetNetEncoderLayer(config)
             for _ in range(config.num_encoder_layers)])
        self.decoder_layers = nn.ModuleList(
            [ProphetNetDecoderLayer(config)
             for _ in range(config.num_decoder_layers)])
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def encode(self, input_ids, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.embed_norm(self.word_embeddings(input_ids) +
                            self.position_embeddings(pos))
        mask = (expand_padding_mask(attention_mask, x.dtype)
                if attention_mask is not None else None)
        for layer in self.encoder_layers:
            x = layer(x, mask)
        return x

    def decode(self, decoder_input_ids, enc, enc_pad_mask=None):
        B, S = decoder_input_ids.shape
        pos = torch.arange(S, device=decoder_input_ids.device)
        base = self.word_embeddings(decoder_input_ids) + \
            self.position_embeddings(pos)
        # predict stream g = base + ngram_embedding[g]
        streams = [base] + [base + self.ngram_embeddings[g]
                            for g in range(self.config.ngram)]
        x = self.embed_norm(torch.cat(streams, dim=1))
        for layer in self.decoder_layers:
            x = layer(x, S, enc, enc_pad_mask)
        # -> main [B,S,H], predict [ngram,B,S,H]
        parts = x.chunk(1 + self.config.ngram, dim=1)
        return parts[0], torch.stack(parts[1:], dim=0)

    def forward(self, input_ids, decoder_input_ids, attention_mask=None):
        enc = self.encode(input_ids, attention_mask)
        enc_mask = (expand_padding_mask(attention_mask, enc.dtype)
                    if attention_mask is not None else None)
        return self.decode(decoder_input_ids, enc, enc_mask)


class ProphetNetForConditionalGeneration(ProphetNetPretrainedModel,
                                         GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: ProphetNetConfig):
        super().__init__(config)
        self.prophetnet = ProphetNetModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=False)
        self.lm_head.weight = self.prophetnet.word_embeddings.weight

    def forward(self, input_ids, decoder_input_ids=None, attention_mask=None,
                labels=None, **kwargs):
        if decoder_input_ids is None and labels is not None:
            start = torch.full((labels.shape[0], 1),
                               self.config.decoder_start_token_id,
                               dtype=labels.dtype, device=labels.device)
            decoder_input_ids = torch.cat(
                [start, labels[:, :-1].clamp(min=0)], dim=1)
        main, predict = self.prophetnet(input_ids, decoder_input_ids,
                                        attention_mask)
        logits = self.lm_head(main)
        if labels is None:
            return logits
        V = self.config.vocab_size
        loss = F.cross_entropy(logits.reshape(-1, V), labels.reshape(-1),
                               ignore_index=-100)
        # n-gram losses: stream g predicts labels shifted left by g
        # (token t+g), eps-weighted (reference predict loss)
        for g in range(self.config.ngram):
            pl = self.lm_head(predict[g])
            tgt = torch.full_like(labels, -100)
            if g + 1 < labels.shape[1]:
                tgt[:, :-(g + 1)] = labels[:, g + 1:]
            lg = F.cross_entropy(pl.reshape(-1, V), tgt.reshape(-1),
                                 ignore_index=-100)
            if not torch.isnan(lg):
                loss = loss + self.config.eps * lg
        return loss, logits
