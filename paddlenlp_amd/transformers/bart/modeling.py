"""BART encoder-decoder family (reference: paddlenlp/transformers/bart/modeling.py).

Post-LN transformer with learned positional embeddings offset past the
padding index (+2, the BART quirk), an embedding LayerNorm, causal decoder
with cross-attention and KV caches, tied LM head with a final-logits bias,
and shift-right label preparation starting from decoder_start_token_id.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig
from ..model_utils import PretrainedModel
from .configuration import BartConfig

__all__ = ["BartModel", "BartEncoder", "BartDecoder",
           "BartForConditionalGeneration"]

ACT = {"gelu": F.gelu, "relu": F.relu, "silu": F.silu}


class BartLearnedPositionalEmbedding(nn.Embedding):
    """BART reserves 2 extra positions: position i uses row i + 2."""

    OFFSET = 2

    def __init__(self, num_positions: int, d_model: int):
        super().__init__(num_positions + self.OFFSET, d_model)

    def forward(self, seq_len: int, past_len: int = 0):
        pos = torch.arange(past_len, past_len + seq_len,
                           device=self.weight.device)
        return super().forward(pos + self.OFFSET)


class BartAttention(nn.Module):
    def __init__(self, d_model: int, num_heads: int):
        super().__init__()
        self.num_heads = num_heads
        self.head_dim = d_model // num_heads
        self.scaling = self.head_dim ** -0.5
        self.q_proj = nn.Linear(d_model, d_model, bias=True)
        self.k_proj = nn.Linear(d_model, d_model, bias=True)
        self.v_proj = nn.Linear(d_model, d_model, bias=True)
        self.out_proj = nn.Linear(d_model, d_model, bias=True)

    def forward(self, x, kv=None, causal=False, past_key_value=None,
                use_cache=False):
        B, S, D = x.shape
        H, hd = self.num_heads, self.head_dim
        q = self.q_proj(x).view(B, S, H, hd).transpose(1, 2)
        if past_key_value is not None and kv is not None:
            k, v = past_key_value  # cross-attention cache
        else:
            src = x if kv is None else kv
            k = self.k_proj(src).view(B, -1, H, hd).transpose(1, 2)
            v = self.v_proj(src).view(B, -1, H, hd).transpose(1, 2)
            if past_key_value is not None:
                k = torch.cat([past_key_value[0], k], dim=2)
                v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None
        Skv = k.shape[2]
        attn_mask = None
        is_causal = causal and S > 1 and S == Skv
        if causal and S > 1 and S != Skv:
            m = torch.full((S, Skv), float("-inf"), device=x.device, dtype=x.dtype)
            attn_mask = m.triu(Skv - S + 1)
        out = F.scaled_dot_product_attention(
            q, k, v, attn_mask=attn_mask, is_causal=is_causal,
            scale=self.scaling)
        out = self.out_proj(out.transpose(1, 2).reshape(B, S, D))
        if use_cache:
            return out, present
        return out


class BartEncoderLayer(nn.Module):
    def __init__(self, config: BartConfig):
        super().__init__()
        d = config.d_model
        self.self_attn = BartAttention(d, config.encoder_attention_heads)
        self.self_attn_layer_norm = nn.LayerNorm(d)
        self.fc1 = nn.Linear(d, config.encoder_ffn_dim)
        self.fc2 = nn.Linear(config.encoder_ffn_dim, d)
        self.final_layer_norm = nn.LayerNorm(d)
        self.act = ACT[config.activation_function]

    def forward(self, x):
        x = self.self_attn_layer_norm(x + self.self_attn(x))
        x = self.final_layer_norm(x + self.fc2(self.act(self.fc1(x))))
        return x


class BartDecoderLayer(nn.Module):
    def __init__(self, config: BartConfig):
        super().__init__()
        d = config.d_model
        self.self_attn = BartAttention(d, config.decoder_attention_heads)
        self.self_attn_layer_norm = nn.LayerNorm(d)
        self.encoder_attn = BartAttention(d, config.decoder_attention_heads)
        self.encoder_attn_layer_norm = nn.LayerNorm(d)
        self.fc1 = nn.Linear(d, config.decoder_ffn_dim)
        self.fc2 = nn.Linear(config.decoder_ffn_dim, d)
        self.final_layer_norm = nn.LayerNorm(d)
        self.act = ACT[config.activation_function]

    def forward(self, x, encoder_out, past_key_value=None, use_cache=False):
        self_past = cross_past = None
        if past_key_value is not None:
            self_past, cross_past = past_key_value
        h = self.self_attn(x, causal=True, past_key_value=self_past,
                           use_cache=use_cache)
        if use_cache:
            h, self_present = h
        x = self.self_attn_layer_norm(x + h)
        h = self.encoder_attn(x, kv=encoder_out, past_key_value=cross_past,
                              use_cache=use_cache)
        cross_present = None
        if use_cache:
            h, cross_present = h
        x = self.encoder_attn_layer_norm(x + h)
        x = self.final_layer_norm(x + self.fc2(self.act(self.fc1(x))))
        if use_cache:
            return x, (self_present, cross_present)
        return x


class BartPretrainedModel(PretrainedModel):
    config_class = BartConfig
    base_model_prefix = "bart"

    def _init_weights(self, module):
        std = self.config.init_std
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, nn.LayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()


class BartEncoder(nn.Module):
    def __init__(self, config: BartConfig, embed_tokens: nn.Embedding):
        super().__init__()
        self.embed_tokens = embed_tokens
        self.embed_positions = BartLearnedPositionalEmbedding(
            config.max_position_embeddings, config.d_model)
        self.layernorm_embedding = nn.LayerNorm(config.d_model)
        self.embed_scale = (math.sqrt(config.d_model)
                            if config.scale_embedding else 1.0)
        self.layers = nn.ModuleList(
            [BartEncoderLayer(config) for _ in range(config.encoder_layers)])

    def forward(self, input_ids):
        x = self.embed_tokens(input_ids) * self.embed_scale
        x = x + self.embed_positions(input_ids.shape[1])
        x = self.layernorm_embedding(x)
        for layer in self.layers:
            x = layer(x)
        return x


class BartDecoder(nn.Module):
    def __init__(self, config: BartConfig, embed_tokens: nn.Embedding):
        super().__init__()
        self.embed_tokens = embed_tokens
        self.embed_positions = BartLearnedPositionalEmbedding(
            config.max_position_embeddings, config.d_model)
        self.layernorm_embedding = nn.LayerNorm(config.d_model)
        self.embed_scale = (math.sqrt(config.d_model)
                            if config.scale_embedding else 1.0)
        self.layers = nn.ModuleList(
            [BartDecoderLayer(config) for _ in range(config.decoder_layers)])

    def forward(self, input_ids, encoder_out, past_key_values=None,
                use_cache=False):
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None \
                and past_key_values[0][0] is not None:
            past_len = past_key_values[0][0][0].shape[2]
        x = self.embed_tokens(input_ids) * self.embed_scale
        x = x + self.embed_positions(input_ids.shape[1], past_len)
        x = self.layernorm_embedding(x)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, encoder_out, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        if use_cache:
            return x, presents
        return x


class BartModel(BartPretrainedModel):
    _tied_weights_keys = ["encoder.embed_tokens.weight",
                          "decoder.embed_tokens.weight"]

    def __init__(self, config: BartConfig):
        super().__init__(config)
        self.shared = nn.Embedding(config.vocab_size, config.d_model,
                                   padding_idx=config.pad_token_id)
        self.encoder = BartEncoder(config, self.shared)
        self.decoder = BartDecoder(config, self.shared)

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


class BartForConditionalGeneration(BartPretrainedModel):
    _tied_weights_keys = ["bart.encoder.embed_tokens.weight",
                          "bart.decoder.embed_tokens.weight",
                          "lm_head.weight"]

    def __init__(self, config: BartConfig):
        super().__init__(config)
        self.bart = BartModel(config)
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        self.lm_head.weight = self.bart.shared.weight
        self.register_buffer("final_logits_bias",
                             torch.zeros(config.vocab_size))
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        self.bart.tie_weights()
        self.lm_head.weight = self.bart.shared.weight

    def get_input_embeddings(self):
        return self.bart.shared

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
        out = self.bart(input_ids, decoder_input_ids, encoder_output,
                        past_key_values, use_cache)
        if use_cache:
            hidden, presents, enc = out
        else:
            hidden, enc = out
            presents = None
        logits = self.lm_head(hidden) + self.final_logits_bias
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]),
                labels.reshape(-1), -100, reduction="mean")
            return (loss, logits) if not use_cache else (loss, logits, presents, enc)
        return logits if not use_cache else (logits, presents, enc)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor,
                 generation_config=None, **kwargs):
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
        enc = self.bart.encoder(input_ids)
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
            nxt = logits[:, -1].float()
            if gen.do_sample:
                nxt = nxt / max(gen.temperature, 1e-6)
                token = torch.multinomial(nxt.softmax(-1), 1).squeeze(-1)
            else:
                token = nxt.argmax(-1)
            token = torch.where(unfinished, token, torch.full_like(token, pad))
            tokens.append(token)
            cur = token[:, None]
            for e in eos:
                unfinished = unfinished & (token != e)
            if not unfinished.any():
                break
        out = torch.stack(tokens, dim=1) if tokens else input_ids.new_zeros(B, 0)
        return out, None
