"""Pegasus family (reference: paddlenlp/transformers/pegasus/modeling.py).

Summarization encoder-decoder: PRE-LN transformer stacks (unlike BART's
post-LN), SINUSOIDAL (non-learned) position embeddings, scaled embeddings,
untied-bias-free LM head over the shared embedding.  Reuses the BART
attention/cache machinery.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig
from ..bart.modeling import BartAttention
from ..configuration_utils import PretrainedConfig
from ..model_utils import PretrainedModel

__all__ = ["PegasusConfig", "PegasusModel", "PegasusForConditionalGeneration"]


class PegasusConfig(PretrainedConfig):
    model_type = "pegasus"

    def __init__(self, vocab_size=96103, d_model=1024, encoder_layers=16,
                 decoder_layers=16, encoder_attention_heads=16,
                 decoder_attention_heads=16, encoder_ffn_dim=4096,
                 decoder_ffn_dim=4096, activation_function="relu",
                 max_position_embeddings=1024, init_std=0.02,
                 pad_token_id=0, eos_token_id=1, decoder_start_token_id=0,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.d_model = d_model
        self.encoder_layers = encoder_layers
        self.decoder_layers = decoder_layers
        self.encoder_attention_heads = encoder_attention_heads
        self.decoder_attention_heads = decoder_attention_heads
        self.encoder_ffn_dim = encoder_ffn_dim
        self.decoder_ffn_dim = decoder_ffn_dim
        self.activation_function = activation_function
        self.max_position_embeddings = max_position_embeddings
        self.init_std = init_std
        self.pad_token_id = pad_token_id
        self.eos_token_id = eos_token_id
        self.decoder_start_token_id = decoder_start_token_id


def sinusoidal_positions(n_pos: int, dim: int) -> torch.Tensor:
    pos = torch.arange(n_pos, dtype=torch.float32)[:, None]
    i = torch.arange(dim // 2, dtype=torch.float32)[None, :]
    angle = pos / (10000.0 ** (2 * i / dim))
    out = torch.zeros(n_pos, dim)
    out[:, 0::2] = torch.sin(angle)
    out[:, 1::2] = torch.cos(angle)
    return out


ACT = {"relu": F.relu, "gelu": F.gelu, "silu": F.silu}


class _PegasusEncoderLayer(nn.Module):
    def __init__(self, c: PegasusConfig):
        super().__init__()
        d = c.d_model
        self.self_attn_layer_norm = nn.LayerNorm(d)
        self.self_attn = BartAttention(d, c.encoder_attention_heads)
        self.final_layer_norm = nn.LayerNorm(d)
        self.fc1 = nn.Linear(d, c.encoder_ffn_dim)
        self.fc2 = nn.Linear(c.encoder_ffn_dim, d)
        self.act = ACT[c.activation_function]

    def forward(self, x):  # pre-LN
        x = x + self.self_attn(self.self_attn_layer_norm(x))
        x = x + self.fc2(self.act(self.fc1(self.final_layer_norm(x))))
        return x


class _PegasusDecoderLayer(nn.Module):
    def __init__(self, c: PegasusConfig):
        super().__init__()
        d = c.d_model
        self.self_attn_layer_norm = nn.LayerNorm(d)
        self.self_attn = BartAttention(d, c.decoder_attention_heads)
        self.encoder_attn_layer_norm = nn.LayerNorm(d)
        self.encoder_attn = BartAttention(d, c.decoder_attention_heads)
        self.final_layer_norm = nn.LayerNorm(d)
        self.fc1 = nn.Linear(d, c.decoder_ffn_dim)
        self.fc2 = nn.Linear(c.decoder_ffn_dim, d)
        self.act = ACT[c.activation_function]

    def forward(self, x, encoder_out, past_key_value=None, use_cache=False):
        self_past = cross_past = None
        if past_key_value is not None:
            self_past, cross_past = past_key_value
        h = self.self_attn(self.self_attn_layer_norm(x), causal=True,
                           past_key_value=self_past, use_cache=use_cache)
        if use_cache:
            h, self_present = h
        x = x + h
        h = self.encoder_attn(self.encoder_attn_layer_norm(x), kv=encoder_out,
                              past_key_value=cross_past, use_cache=use_cache)
        cross_present = None
        if use_cache:
            h, cross_present = h
        x = x + h
        x = x + self.fc2(self.act(self.fc1(self.final_layer_norm(x))))
        if use_cache:
            return x, (self_present, cross_present)
        return x


class PegasusPretrainedModel(PretrainedModel):
    config_class = PegasusConfig
    base_model_prefix = "pegasus"

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


class _SinusoidalPositions(nn.Module):
    """Lazily materialized (meta-device-safe, like LlamaRotaryEmbedding)."""

    def __init__(self, n_pos: int, dim: int):
        super().__init__()
        self.n_pos, self.dim = n_pos, dim
        self.register_buffer("table", torch.empty(0), persistent=False)

    def forward(self, start: int, length: int, ref: torch.Tensor) -> torch.Tensor:
        if self.table.numel() == 0 or self.table.device != ref.device:
            self.table = sinusoidal_positions(self.n_pos, self.dim).to(
                device=ref.device, dtype=ref.dtype)
        return self.table[start:start + length]


class _PegasusEncoder(nn.Module):
    def __init__(self, c: PegasusConfig, embed):
        super().__init__()
        self.embed_tokens = embed
        self.pos = _SinusoidalPositions(c.max_position_embeddings, c.d_model)
        self.scale = math.sqrt(c.d_model)
        self.layers = nn.ModuleList(
            [_PegasusEncoderLayer(c) for _ in range(c.encoder_layers)])
        self.layer_norm = nn.LayerNorm(c.d_model)

    def forward(self, input_ids):
        S = input_ids.shape[1]
        x = self.embed_tokens(input_ids) * self.scale
        x = x + self.pos(0, S, x)
        for layer in self.layers:
            x = layer(x)
        return self.layer_norm(x)


class _PegasusDecoder(nn.Module):
    def __init__(self, c: PegasusConfig, embed):
        super().__init__()
        self.embed_tokens = embed
        self.pos = _SinusoidalPositions(c.max_position_embeddings, c.d_model)
        self.scale = math.sqrt(c.d_model)
        self.layers = nn.ModuleList(
            [_PegasusDecoderLayer(c) for _ in range(c.decoder_layers)])
        self.layer_norm = nn.LayerNorm(c.d_model)

    def forward(self, input_ids, encoder_out, past_key_values=None,
                use_cache=False):
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None \
                and past_key_values[0][0] is not None:
            past_len = past_key_values[0][0][0].shape[2]
        S = input_ids.shape[1]
        x = self.embed_tokens(input_ids) * self.scale
        x = x + self.pos(past_len, S, x)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, encoder_out, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.layer_norm(x)
        if use_cache:
            return x, presents
        return x


class PegasusModel(PegasusPretrainedModel):
    _tied_weights_keys = ["encoder.embed_tokens.weight",
                          "decoder.embed_tokens.weight"]

    def __init__(self, config: PegasusConfig):
        super().__init__(config)
        self.shared = nn.Embedding(config.vocab_size, config.d_model,
                                   padding_idx=config.pad_token_id)
        self.encoder = _PegasusEncoder(config, self.shared)
        self.decoder = _PegasusDecoder(config, self.shared)

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


class PegasusForConditionalGeneration(PegasusPretrainedModel):
    _tied_weights_keys = ["pegasus.encoder.embed_tokens.weight",
                          "pegasus.decoder.embed_tokens.weight",
                          "lm_head.weight"]

    def __init__(self, config: PegasusConfig):
        super().__init__(config)
        self.pegasus = PegasusModel(config)
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        self.lm_head.weight = self.pegasus.shared.weight
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        self.pegasus.tie_weights()
        self.lm_head.weight = self.pegasus.shared.weight

    def get_input_embeddings(self):
        return self.pegasus.shared

    def _shift_right(self, labels):
        start = torch.full_like(labels[:, :1], self.config.decoder_start_token_id)
        shifted = torch.cat([start, labels[:, :-1]], dim=1)
        return shifted.masked_fill(shifted == -100, self.config.pad_token_id)

    def forward(self, input_ids=None, decoder_input_ids=None, labels=None,
                encoder_output=None, past_key_values=None, use_cache=False):
        if decoder_input_ids is None and labels is not None:
            decoder_input_ids = self._shift_right(labels)
        out = self.pegasus(input_ids, decoder_input_ids, encoder_output,
                           past_key_values, use_cache)
        if use_cache:
            hidden, presents, enc = out
        else:
            hidden, enc = out
            presents = None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]),
                labels.reshape(-1), -100, reduction="mean")
            return (loss, logits) if not use_cache else (loss, logits, presents, enc)
        return logits if not use_cache else (logits, presents, enc)

    @torch.no_grad()
    def generate(self, input_ids, generation_config=None, **kwargs):
        gen = generation_config or self.generation_config
        for k, v in kwargs.items():
            if hasattr(gen, k):
                setattr(gen, k, v)
        from ...generation.seq2seq_utils import seq2seq_beam_search

        eos = (gen.eos_ids() or [self.config.eos_token_id])[0]
        pad = gen.pad_token_id if gen.pad_token_id is not None \
            else self.config.pad_token_id
        if gen.num_beams > 1:
            return seq2seq_beam_search(
                self, input_ids, gen,
                start_token_id=self.config.decoder_start_token_id,
                eos_token_id=eos, pad_token_id=pad)
        # greedy/sample loop
        B = input_ids.shape[0]
        device = input_ids.device
        enc = self.pegasus.encoder(input_ids)
        cur = torch.full((B, 1), self.config.decoder_start_token_id,
                         dtype=torch.long, device=device)
        past = None
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
            unfinished = unfinished & (token != eos)
            if not unfinished.any():
                break
        out = torch.stack(tokens, dim=1) if tokens else input_ids.new_zeros(B, 0)
        return out, None
