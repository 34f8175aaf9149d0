"""Blenderbot family (reference: paddlenlp/transformers/blenderbot/modeling.py).

Open-domain dialogue seq2seq: PRE-LN encoder/decoder (normalize_before=True)
with a final stack layernorm, LEARNED positions WITHOUT bart's +2 offset,
scaled embeddings, tied LM head.  Both towers reuse the pegasus pre-LN layer
classes (identical shape); blenderbot_small subclasses with post-LN layers.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn

from ... import ops
from ...generation import GenerationConfig
from ..configuration_utils import PretrainedConfig
from ..model_utils import PretrainedModel
from ..pegasus.modeling import _PegasusDecoderLayer, _PegasusEncoderLayer

__all__ = ["BlenderbotConfig", "BlenderbotModel",
           "BlenderbotForConditionalGeneration"]


class BlenderbotConfig(PretrainedConfig):
    model_type = "blenderbot"

    def __init__(self, vocab_size=8008, d_model=2560, encoder_layers=2,
                 decoder_layers=24, encoder_attention_heads=32,
                 decoder_attention_heads=32, encoder_ffn_dim=10240,
                 decoder_ffn_dim=10240, activation_function="gelu",
                 max_position_embeddings=128, init_std=0.02,
                 scale_embedding=True, normalize_before=True,
                 pad_token_id=0, bos_token_id=1, eos_token_id=2,
                 decoder_start_token_id=1, **kwargs):
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
        self.scale_embedding = scale_embedding
        self.normalize_before = normalize_before
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.decoder_start_token_id = decoder_start_token_id


class BlenderbotPretrainedModel(PretrainedModel):
    config_class = BlenderbotConfig
    base_model_prefix = "blenderbot"

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


class _BlenderbotEncoder(nn.Module):
    encoder_layer_cls = _PegasusEncoderLayer

    def __init__(self, c, embed):
        super().__init__()
        self.embed_tokens = embed
        self.embed_positions = nn.Embedding(c.max_position_embeddings, c.d_model)
        self.scale = math.sqrt(c.d_model) if c.scale_embedding else 1.0
        self.layers = nn.ModuleList(
            [self.encoder_layer_cls(c) for _ in range(c.encoder_layers)])
        self.layer_norm = nn.LayerNorm(c.d_model) if c.normalize_before else None

    def forward(self, input_ids):
        pos = torch.arange(input_ids.shape[1], device=input_ids.device)
        x = self.embed_tokens(input_ids) * self.scale + self.embed_positions(pos)
        for layer in self.layers:
            x = layer(x)
        return self.layer_norm(x) if self.layer_norm is not None else x


class _BlenderbotDecoder(nn.Module):
    decoder_layer_cls = _PegasusDecoderLayer

    def __init__(self, c, embed):
        super().__init__()
        self.embed_tokens = embed
        self.embed_positions = nn.Embedding(c.max_position_embeddings, c.d_model)
        self.scale = math.sqrt(c.d_model) if c.scale_embedding else 1.0
        self.layers = nn.ModuleList(
            [self.decoder_layer_cls(c) for _ in range(c.decoder_layers)])
        self.layer_norm = nn.LayerNorm(c.d_model) if c.normalize_before else None

    def forward(self, input_ids, encoder_out, past_key_values=None,
                use_cache=False):
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None \
                and past_key_values[0][0] is not None:
            past_len = past_key_values[0][0][0].shape[2]
        pos = torch.arange(past_len, past_len + input_ids.shape[1],
                           device=input_ids.device)
        x = self.embed_tokens(input_ids) * self.scale + self.embed_positions(pos)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, encoder_out, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        if self.layer_norm is not None:
            x = self.layer_norm(x)
        if use_cache:
            return x, presents
        return x


class BlenderbotModel(BlenderbotPretrainedModel):
    encoder_cls = _BlenderbotEncoder
    decoder_cls = _BlenderbotDecoder
    _tied_weights_keys = ["encoder.embed_tokens.weight",
                          "decoder.embed_tokens.weight"]

    def __init__(self, config):
        super().__init__(config)
        self.shared = nn.Embedding(config.vocab_size, config.d_model,
                                   padding_idx=config.pad_token_id)
        self.encoder = self.encoder_cls(config, self.shared)
        self.decoder = self.decoder_cls(config, self.shared)

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


class BlenderbotForConditionalGeneration(BlenderbotPretrainedModel):
    model_cls = BlenderbotModel
    base_attr = "blenderbot"
    _tied_weights_keys = ["blenderbot.encoder.embed_tokens.weight",
                          "blenderbot.decoder.embed_tokens.weight",
                          "lm_head.weight"]

    def __init__(self, config):
        super().__init__(config)
        setattr(self, self.base_attr, self.model_cls(config))
        self.lm_head = nn.Linear(config.d_model, config.vocab_size, bias=False)
        self.lm_head.weight = self.base.shared.weight
        self.generation_config = GenerationConfig.from_model_config(config)

    @property
    def base(self):
        return getattr(self, self.base_attr)

    def tie_weights(self):
        self.base.tie_weights()
        self.lm_head.weight = self.base.shared.weight

    def get_input_embeddings(self):
        return self.base.shared

    def _shift_right(self, labels):
        start = torch.full_like(labels[:, :1], self.config.decoder_start_token_id)
        shifted = torch.cat([start, labels[:, :-1]], dim=1)
        return shifted.masked_fill(shifted == -100, self.config.pad_token_id)

    def forward(self, input_ids=None, decoder_input_ids=None, labels=None,
                encoder_output=None, past_key_values=None, use_cache=False):
        if decoder_input_ids is None and labels is not None:
            decoder_input_ids = self._shift_right(labels)
        out = self.base(input_ids, decoder_input_ids, encoder_output,
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
        B = input_ids.shape[0]
        enc = self.base.encoder(input_ids)
        cur = torch.full((B, 1), self.config.decoder_start_token_id,
                         dtype=torch.long, device=input_ids.device)
        past = None
        unfinished = torch.ones(B, dtype=torch.bool, device=input_ids.device)
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
