"""T5 encoder-decoder family (reference: paddlenlp/transformers/t5/modeling.py).

Relative-position-bias attention (bias computed by the first layer of each
stack and shared down the stack), RMS-style T5LayerNorm through the fused
rms_norm op seam, unscaled attention (T5 folds 1/sqrt(d) into the weights),
optional gated activations, tied embeddings with the d_model**-0.5 logits
rescale, and a cached greedy/sampling seq2seq generate loop.
"""
from __future__ import annotations

import math
from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig
from ..model_utils import PretrainedModel
from .configuration import T5Config

__all__ = ["T5Model", "T5EncoderModel", "T5ForConditionalGeneration"]

ACT = {"relu": F.relu, "gelu": lambda x: F.gelu(x, approximate="tanh"),
       "silu": F.silu}


class T5LayerNorm(nn.Module):
    """RMS norm without bias or mean subtraction (T5 style)."""

    def __init__(self, d_model: int, eps: float = 1e-6):
        super().__init__()
        self.weight = nn.Parameter(torch.ones(d_model))
        self.eps = eps

    def forward(self, x):
        return ops.rms_norm(x, self.weight, self.eps)


def relative_position_bucket(rel_pos: torch.Tensor, bidirectional: bool,
                             num_buckets: int, max_distance: int) -> torch.Tensor:
    """T5's log-bucketed relative positions (reference _relative_position_bucket)."""
    ret = torch.zeros_like(rel_pos)
    if bidirectional:
        num_buckets //= 2
        ret = ret + (rel_pos > 0).long() * num_buckets
        n = rel_pos.abs()
    else:
        n = (-rel_pos).clamp(min=0)
    max_exact = num_buckets // 2
    is_small = n < max_exact
    large = max_exact + (
        torch.log(n.float().clamp(min=1) / max_exact)
        / math.log(max_distance / max_exact) * (num_buckets - max_exact)
    ).long()
    large = torch.minimum(large, torch.full_like(large, num_buckets - 1))
    return ret + torch.where(is_small, n, large)


class T5Attention(nn.Module):
    def __init__(self, config: T5Config, has_relative_attention_bias: bool,
                 bidirectional: bool):
        super().__init__()
        self.num_heads = config.num_heads
        self.d_kv = config.d_kv
        inner = config.num_heads * config.d_kv
        self.q = nn.Linear(config.d_model, inner, bias=False)
        self.k = nn.Linear(config.d_model, inner, bias=False)
        self.v = nn.Linear(config.d_model, inner, bias=False)
        self.o = nn.Linear(inner, config.d_model, bias=False)
        self.bidirectional = bidirectional
        self.has_relative_attention_bias = has_relative_attention_bias
        if has_relative_attention_bias:
            self.relative_attention_bias = nn.Embedding(
                config.relative_attention_num_buckets, config.num_heads)
        self.num_buckets = config.relative_attention_num_buckets
        self.max_distance = config.relative_attention_max_distance

    def compute_bias(self, q_len: int, kv_len: int, device) -> torch.Tensor:
        """[1, H, q_len, kv_len] additive bias."""
        ctx = torch.arange(q_len, device=device)[:, None]
        mem = torch.arange(kv_len, device=device)[None, :]
        buckets = relative_position_bucket(
            mem - ctx, self.bidirectional, self.num_buckets, self.max_distance)
        return self.relative_attention_bias(buckets).permute(2, 0, 1).unsqueeze(0)

    def forward(self, x, kv=None, position_bias=None, causal=False,
                past_key_value=None, use_cache=False):
        B, S, _ = x.shape
        H, D = self.num_heads, self.d_kv
        q = self.q(x).view(B, S, H, D).transpose(1, 2)          # [B,H,S,D]
        kv_src = x if kv is None else kv
        if past_key_value is not None and kv is not None:
            # cross-attention cache: encoder K/V computed once
            k, v = past_key_value
        else:
            k = self.k(kv_src).view(B, -1, H, D).transpose(1, 2)
            v = self.v(kv_src).view(B, -1, H, D).transpose(1, 2)
            if past_key_value is not None:  # self-attention decode cache
                k = torch.cat([past_key_value[0], k], dim=2)
                v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None

        mask = position_bias
        if causal and S > 1:
            cmask = torch.full((S, k.shape[2]), float("-inf"), device=x.device)
            cmask = cmask.triu(k.shape[2] - S + 1)
            mask = cmask if mask is None else mask + cmask
        # T5 attention is UNscaled: the 1/sqrt(d) lives in the init
        out = F.scaled_dot_product_attention(q, k, v, attn_mask=mask, scale=1.0)
        out = self.o(out.transpose(1, 2).reshape(B, S, H * D))
        return out, present


class T5FeedForward(nn.Module):
    def __init__(self, config: T5Config):
        super().__init__()
        self.act = ACT[config.dense_act_fn]
        self.gated = config.is_gated_act
        if self.gated:
            self.wi_0 = nn.Linear(config.d_model, config.d_ff, bias=False)
            self.wi_1 = nn.Linear(config.d_model, config.d_ff, bias=False)
        else:
            self.wi = nn.Linear(config.d_model, config.d_ff, bias=False)
        self.wo = nn.Linear(config.d_ff, config.d_model, bias=False)

    def forward(self, x):
        if self.gated:
            h = self.act(self.wi_0(x)) * self.wi_1(x)
        else:
            h = self.act(self.wi(x))
        return self.wo(h)


class T5Block(nn.Module):
    def __init__(self, config: T5Config, is_decoder: bool, has_bias: bool):
        super().__init__()
        self.is_decoder = is_decoder
        self.self_attn = T5Attention(config, has_bias, bidirectional=not is_decoder)
        self.self_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)
        if is_decoder:
            self.cross_attn = T5Attention(config, False, bidirectional=True)
            self.cross_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)
        self.ff = T5FeedForward(config)
        self.ff_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)

    def forward(self, x, encoder_out=None, position_bias=None,
                past_key_value=None, use_cache=False):
        self_past = cross_past = None
        if past_key_value is not None:
            self_past, cross_past = past_key_value
        h, self_present = self.self_attn(
            self.self_norm(x), position_bias=position_bias,
            causal=self.is_decoder, past_key_value=self_past, use_cache=use_cache)
        x = x + h
        cross_present = None
        if self.is_decoder:
            h, cross_present = self.cross_attn(
                self.cross_norm(x), kv=encoder_out,
                past_key_value=cross_past, use_cache=use_cache)
            x = x + h
        x = x + self.ff(self.ff_norm(x))
        present = (self_present, cross_present) if use_cache else None
        return x, present


class T5Stack(nn.Module):
    def __init__(self, config: T5Config, embed: nn.Embedding, is_decoder: bool):
        super().__init__()
        self.embed_tokens = embed
        self.is_decoder = is_decoder
        n = config.num_decoder_layers if is_decoder else config.num_layers
        self.blocks = nn.ModuleList(
            [T5Block(config, is_decoder, has_bias=(i == 0)) for i in range(n)])
        self.final_norm = T5LayerNorm(config.d_model, config.layer_norm_epsilon)

    def forward(self, input_ids, encoder_out=None, past_key_values=None,
                use_cache=False):
        x = self.embed_tokens(input_ids)
        S = input_ids.shape[1]
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None \
                and past_key_values[0][0] is not None:
            past_len = past_key_values[0][0][0].shape[2]
        kv_len = past_len + S
        # the first block owns the bias table; queries sit at the cache tail
        bias = self.blocks[0].self_attn.compute_bias(
            kv_len, kv_len, x.device)[:, :, past_len:, :]
        presents = [] if use_cache else None
        for i, block in enumerate(self.blocks):
            past = past_key_values[i] if past_key_values is not None else None
            x, present = block(x, encoder_out, bias, past, use_cache)
            if use_cache:
                presents.append(present)
        x = self.final_norm(x)
        if use_cache:
            return x, presents
        return x


class T5PretrainedModel(PretrainedModel):
    config_class = T5Config
    base_model_prefix = "t5"

    def _init_weights(self, module):
        factor = self.config.initializer_factor
        d_model = self.config.d_model
        if isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=factor)
        elif isinstance(module, T5LayerNorm):
            module.weight.data.fill_(1.0)
        elif isinstance(module, nn.Linear):
            # T5 folds the attention scale into the init
            module.weight.data.normal_(mean=0.0, std=factor * d_model ** -0.5)


class T5Model(T5PretrainedModel):
    # encoder/decoder stacks alias the shared embedding table
    _tied_weights_keys = ["encoder.embed_tokens.weight",
                          "decoder.embed_tokens.weight"]

    def __init__(self, config: T5Config):
        super().__init__(config)
        self.shared = nn.Embedding(config.vocab_size, config.d_model)
        self.encoder = T5Stack(config, self.shared, is_decoder=False)
        self.decoder = T5Stack(config, self.shared, is_decoder=True)

    def tie_weights(self):
        self.encoder.embed_tokens.weight = self.shared.weight
        self.decoder.embed_tokens.weight = self.shared.weight

    def get_input_embeddings(self):
        return self.shared

    def forward(self, input_ids, decoder_input_ids, encoder_output=None,
                past_key_values=None, use_cache=False):
        if encoder_output is None:
            encoder_output = self.encoder(input_ids)
        out = self.decoder(decoder_input_ids, encoder_output,
                           past_key_values, use_cache)
        if use_cache:
            hidden, presents = out
            return hidden, presents, encoder_output
        return out, encoder_output


class T5EncoderModel(T5PretrainedModel):
    _tied_weights_keys = ["encoder.embed_tokens.weight"]

    def __init__(self, config: T5Config):
        super().__init__(config)
        self.shared = nn.Embedding(config.vocab_size, config.d_model)
        self.encoder = T5Stack(config, self.shared, is_decoder=False)

    def tie_weights(self):
        self.encoder.embed_tokens.weight = self.shared.weight

    def get_input_embeddings(self):
        return self.shared

    def forward(self, input_ids):
        return self.encoder(input_ids)


class T5ForConditionalGeneration(T5PretrainedModel):
    _tied_weights_keys = ["t5.encoder.embed_tokens.weight",
                          "t5.decoder.embed_tokens.weight"]

    def __init__(self, config: T5Config):
        super().__init__(config)
        self.t5 = T5Model(config)
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.t5.shared.weight
            # instance attr shadows the class list: lm_head is only dropped
            # from checkpoints when it actually aliases the embedding
            self._tied_weights_keys = self._tied_weights_keys + ["lm_head.weight"]
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        self.t5.tie_weights()
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.t5.shared.weight

    def get_input_embeddings(self):
        return self.t5.shared

    def get_output_embeddings(self):
        return self.lm_head

    def _shift_right(self, labels: torch.Tensor) -> torch.Tensor:
        start = torch.full_like(labels[:, :1], self.config.decoder_start_token_id)
        shifted = torch.cat([start, labels[:, :-1]], dim=1)
        return shifted.masked_fill(shifted == -100, self.config.pad_token_id)

    def forward(self, input_ids=None, decoder_input_ids=None, labels=None,
                encoder_output=None, past_key_values=None, use_cache=False):
        if decoder_input_ids is None and labels is not None:
            decoder_input_ids = self._shift_right(labels)
        out = self.t5(input_ids, decoder_input_ids, encoder_output,
                      past_key_values, use_cache)
        if use_cache:
            hidden, presents, enc = out
        else:
            hidden, enc = out
            presents = None
        if self.config.tie_word_embeddings:
            hidden = hidden * (self.config.d_model ** -0.5)
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]),
                labels.reshape(-1), -100, reduction="mean")
            return (loss, logits) if not use_cache else (loss, logits, presents, enc)
        return logits if not use_cache else (logits, presents, enc)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor,
                 generation_config: Optional[GenerationConfig] = None, **kwargs):
        """Cached seq2seq greedy/sampling decode (encoder runs once)."""
        gen = generation_config or self.generation_config
        for k, v in kwargs.items():
            if hasattr(gen, k):
                setattr(gen, k, v)
        if gen.num_beams > 1:
            from ...generation.seq2seq_utils import seq2seq_beam_search

            eos = (gen.eos_ids() or [self.config.eos_token_id])[0]
            pad = gen.pad_token_id if gen.pad_token_id is not None \
                else self.config.pad_token_id
            return seq2seq_beam_search(
                self, input_ids, gen,
                start_token_id=self.config.decoder_start_token_id,
                eos_token_id=eos, pad_token_id=pad)
        B = input_ids.shape[0]
        device = input_ids.device
        enc = self.t5.encoder(input_ids)
        cur = torch.full((B, 1), self.config.decoder_start_token_id,
                         dtype=torch.long, device=device)
        past = None
        eos = gen.eos_ids() or [self.config.eos_token_id]
        pad = gen.pad_token_id if gen.pad_token_id is not None else self.config.pad_token_id
        unfinished = torch.ones(B, dtype=torch.bool, device=device)
        tokens = []
        for _ in range(gen.max_new_tokens):
            logits, past, _ = self.forward(
                decoder_input_ids=cur, encoder_output=enc,
                past_key_values=past, use_cache=True)
            next_logits = logits[:, -1].float()
            if gen.do_sample:
                next_logits = next_logits / max(gen.temperature, 1e-6)
                token = torch.multinomial(next_logits.softmax(-1), 1).squeeze(-1)
            else:
                token = next_logits.argmax(-1)
            token = torch.where(unfinished, token, torch.full_like(token, pad))
            tokens.append(token)
            cur = token[:, None]
            for e in eos:
                unfinished = unfinished & (token != e)
            if not unfinished.any():
                break
        out = torch.stack(tokens, dim=1) if tokens else input_ids.new_zeros(B, 0)
        return out, None
