"""SpeechT5 (reference: paddlenlp/transformers/speecht5/modeling.py).

Unified speech/text encoder-decoder: ONE shared transformer backbone
with modality-specific pre/post-nets —
- speech encoder prenet: strided conv feature encoder over the raw
  waveform (GroupNorm first layer, reference :507-560) + feature
  projection + positional conv embedding (:424-450);
- text encoder prenet: scaled sinusoidal positions over embeddings;
- speech decoder prenet: always-dropout MLP over mel frames (:709-760);
- speech decoder postnet: frame projection + stop logits + residual
  conv refinement (:761 region); text decoder postnet: tied LM head.
Heads: SpeechT5ForSpeechToText (ASR) and SpeechT5ForTextToSpeech (TTS).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["SpeechT5Config", "SpeechT5Model", "SpeechT5ForSpeechToText",
           "SpeechT5ForTextToSpeech"]


class SpeechT5Config(PretrainedConfig):
    model_type = "speecht5"

    def __init__(self, vocab_size=81, hidden_size=768,
                 encoder_layers=12, decoder_layers=6,
                 num_attention_heads=12, intermediate_size=3072,
                 conv_dim=(512, 512, 512, 512, 512, 512, 512),
                 conv_stride=(5, 2, 2, 2, 2, 2, 2),
                 conv_kernel=(10, 3, 3, 3, 3, 2, 2),
                 num_mel_bins=80, reduction_factor=2,
                 speech_decoder_prenet_layers=2,
                 speech_decoder_prenet_units=256,
                 speech_decoder_prenet_dropout=0.5,
                 speech_decoder_postnet_layers=5,
                 speech_decoder_postnet_units=256,
                 speech_decoder_postnet_kernel=5,
                 positional_conv_kernel=128, positional_conv_groups=16,
                 hidden_dropout=0.1, max_position_embeddings=1024,
                 initializer_range=0.02, layer_norm_eps=1e-5,
                 pad_token_id=1, bos_token_id=0, eos_token_id=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.encoder_layers = encoder_layers
        self.decoder_layers = decoder_layers
        self.num_hidden_layers = encoder_layers + decoder_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.conv_dim = list(conv_dim)
        self.conv_stride = list(conv_stride)
        self.conv_kernel = list(conv_kernel)
        self.num_mel_bins = num_mel_bins
        self.reduction_factor = reduction_factor
        self.speech_decoder_prenet_layers = speech_decoder_prenet_layers
        self.speech_decoder_prenet_units = speech_decoder_prenet_units
        self.speech_decoder_prenet_dropout = speech_decoder_prenet_dropout
        self.speech_decoder_postnet_layers = speech_decoder_postnet_layers
        self.speech_decoder_postnet_units = speech_decoder_postnet_units
        self.speech_decoder_postnet_kernel = speech_decoder_postnet_kernel
        self.positional_conv_kernel = positional_conv_kernel
        self.positional_conv_groups = positional_conv_groups
        self.hidden_dropout = hidden_dropout
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


def _sinusoid(S, h, device, dtype):
    pos = torch.arange(S, device=device).float()
    inv = 1.0 / (10000 ** (torch.arange(0, h, 2, device=device).float() / h))
    ang = torch.outer(pos, inv)
    pe = torch.zeros(S, h, device=device)
    pe[:, 0::2] = ang.sin()
    pe[:, 1::2] = ang.cos()
    return pe.to(dtype)


class FeatureEncoder(nn.Module):
    """Strided conv stack over raw audio (reference :507-560)."""

    def __init__(self, config: SpeechT5Config):
        super().__init__()
        layers = []
        cin = 1
        for i, (dim, st, ks) in enumerate(zip(
                config.conv_dim, config.conv_stride, config.conv_kernel)):
            conv = nn.Conv1d(cin, dim, ks, stride=st, bias=False)
            norm = (nn.GroupNorm(dim, dim) if i == 0 else None)
            layers.append(nn.ModuleList([conv, norm]))
            cin = dim
        self.layers = nn.ModuleList(layers)

    def forward(self, input_values):          # [B, T]
        x = input_values[:, None]             # [B, 1, T]
        for conv, norm in self.layers:
            x = conv(x)
            if norm is not None:
                x = norm(x)
            x = F.gelu(x)
        return x.transpose(1, 2)              # [B, frames, C]


class SpeechEncoderPrenet(nn.Module):
    def __init__(self, config: SpeechT5Config):
        super().__init__()
        self.feature_encoder = FeatureEncoder(config)
        self.norm = nn.LayerNorm(config.conv_dim[-1],
                                 eps=config.layer_norm_eps)
        self.projection = nn.Linear(config.conv_dim[-1], config.hidden_size)
        # positional conv embedding (:424-450)
        self.pos_conv = nn.Conv1d(
            config.hidden_size, config.hidden_size,
            config.positional_conv_kernel,
            padding=config.positional_conv_kernel // 2,
            groups=config.positional_conv_groups)

    def forward(self, input_values):
        x = self.projection(self.norm(self.feature_encoder(input_values)))
        pos = self.pos_conv(x.transpose(1, 2))[:, :, :x.shape[1]]
        return x + F.gelu(pos).transpose(1, 2)


class TextEncoderPrenet(nn.Module):
    def __init__(self, config: SpeechT5Config, embed):
        super().__init__()
        self.embed = embed
        self.scale = math.sqrt(config.hidden_size)

    def forward(self, input_ids):
        x = self.embed(input_ids) * self.scale
        return x + _sinusoid(input_ids.shape[1], x.shape[-1],
                             x.device, x.dtype)


class SpeechDecoderPrenet(nn.Module):
    """MLP over mel frames; dropout stays ON at eval (the TTS diversity
    trick, reference :745)."""

    def __init__(self, config: SpeechT5Config):
        super().__init__()
        u = config.speech_decoder_prenet_units
        dims = [config.num_mel_bins] + \
            [u] * config.speech_decoder_prenet_layers
        self.layers = nn.ModuleList(
            [nn.Linear(dims[i], dims[i + 1])
             for i in range(len(dims) - 1)])
        self.final = nn.Linear(u, config.hidden_size)
        self.p = config.speech_decoder_prenet_dropout

    def forward(self, mel):                   # [B, T, mel]
        x = mel
        for lin in self.layers:
            x = F.dropout(F.relu(lin(x)), self.p, training=True)
        x = self.final(x)
        return x + _sinusoid(x.shape[1], x.shape[-1], x.device, x.dtype)


class SpeechDecoderPostnet(nn.Module):
    def __init__(self, config: SpeechT5Config):
        super().__init__()
        mel, r = config.num_mel_bins, config.reduction_factor
        self.feat_out = nn.Linear(config.hidden_size, mel * r)
        self.prob_out = nn.Linear(config.hidden_size, r)
        u, ks = (config.speech_decoder_postnet_units,
                 config.speech_decoder_postnet_kernel)
        convs = []
        for i in range(config.speech_decoder_postnet_layers):
            cin = mel if i == 0 else u
            cout = mel if i == config.speech_decoder_postnet_layers - 1 else u
            convs.append(nn.Sequential(
                nn.Conv1d(cin, cout, ks, padding=ks // 2, bias=False),
                nn.BatchNorm1d(cout)))
        self.postnet = nn.ModuleList(convs)

    def forward(self, h):
        B, T, _ = h.shape
        mel = self.feat_out(h).reshape(B, -1, self.feat_out.out_features //
                                       self.prob_out.out_features)
        stop = self.prob_out(h).reshape(B, -1)
        x = mel.transpose(1, 2)
        for i, conv in enumerate(self.postnet):
            x = conv(x)
            if i < len(self.postnet) - 1:
                x = torch.tanh(x)
        return mel, mel + x.transpose(1, 2), stop


class _Layer(nn.Module):
    def __init__(self, config, cross=False):
        super().__init__()
        h = config.hidden_size
        self.nh, self.dh = config.num_attention_heads, config.head_dim
        self.qkv = nn.Linear(h, 3 * h)
        self.out = nn.Linear(h, h)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.cross = None
        if cross:
            self.cq = nn.Linear(h, h)
            self.ckv = nn.Linear(h, 2 * h)
            self.cout = nn.Linear(h, h)
            self.cross_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
            self.cross = True
        self.ff_in = nn.Linear(h, config.intermediate_size)
        self.ff_out = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)

    def _sa(self, x, causal):
        B, S, H = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        shp = (B, S, self.nh, self.dh)
        o = F.scaled_dot_product_attention(
            q.view(shp).transpose(1, 2), k.view(shp).transpose(1, 2),
            v.view(shp).transpose(1, 2), is_causal=causal)
        return self.out(o.transpose(1, 2).reshape(B, S, H))

    def forward(self, x, enc=None, causal=False):
        x = self.attn_norm(x + self._sa(x, causal))
        if self.cross is not None and enc is not None:
            B, S, H = x.shape
            Se = enc.shape[1]
            q = self.cq(x).view(B, S, self.nh, self.dh).transpose(1, 2)
            k, v = self.ckv(enc).chunk(2, dim=-1)
            k = k.view(B, Se, self.nh, self.dh).transpose(1, 2)
            v = v.view(B, Se, self.nh, self.dh).transpose(1, 2)
            o = F.scaled_dot_product_attention(q, k, v)
            x = self.cross_norm(
                x + self.cout(o.transpose(1, 2).reshape(B, S, H)))
        return self.ff_norm(x + self.ff_out(F.gelu(self.ff_in(x))))


class SpeechT5PretrainedModel(PretrainedModel):
    config_class = SpeechT5Config
    base_model_prefix = "speecht5"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class SpeechT5Model(SpeechT5PretrainedModel):
    """Shared backbone; prenets select the modality per head model."""

    def __init__(self, config: SpeechT5Config):
        super().__init__(config)
        self.shared = nn.Embedding(config.vocab_size, config.hidden_size,
                                   padding_idx=config.pad_token_id)
        self.speech_encoder_prenet = SpeechEncoderPrenet(config)
        self.text_encoder_prenet = TextEncoderPrenet(config, self.shared)
        self.speech_decoder_prenet = SpeechDecoderPrenet(config)
        self.encoder = nn.ModuleList(
            [_Layer(config) for _ in range(config.encoder_layers)])
        self.decoder = nn.ModuleList(
            [_Layer(config, cross=True)
             for _ in range(config.decoder_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.shared

    def encode(self, x):
        for layer in self.encoder:
            x = layer(x)
        return x

    def decode(self, x, enc):
        for layer in self.decoder:
            x = layer(x, enc, causal=True)
        return x


class SpeechT5ForSpeechToText(SpeechT5PretrainedModel):
    _tied_weights_keys = ["text_decoder_postnet.weight"]

    def __init__(self, config: SpeechT5Config):
        super().__init__(config)
        self.speecht5 = SpeechT5Model(config)
        self.text_decoder_postnet = nn.Linear(config.hidden_size,
                                              config.vocab_size, bias=False)
        self.text_decoder_postnet.weight = self.speecht5.shared.weight

    def forward(self, input_values, decoder_input_ids=None, labels=None):
        if decoder_input_ids is None and labels is not None:
            bos = torch.full((labels.shape[0], 1), self.config.bos_token_id,
                             dtype=labels.dtype, device=labels.device)
            decoder_input_ids = torch.cat([bos, labels[:, :-1].clamp(min=0)],
                                          dim=1)
        enc = self.speecht5.encode(
            self.speecht5.speech_encoder_prenet(input_values))
        dec_in = self.speecht5.text_encoder_prenet(decoder_input_ids)
        logits = self.text_decoder_postnet(self.speecht5.decode(dec_in, enc))
        if labels is not None:
            loss = F.cross_entropy(
                logits.reshape(-1, self.config.vocab_size),
                labels.reshape(-1), ignore_index=-100)
            return loss, logits
        return logits


class SpeechT5ForTextToSpeech(SpeechT5PretrainedModel):
    def __init__(self, config: SpeechT5Config):
        super().__init__(config)
        self.speecht5 = SpeechT5Model(config)
        self.speech_decoder_postnet = SpeechDecoderPostnet(config)

    def forward(self, input_ids, decoder_mel=None, labels=None,
                stop_labels=None):
        """labels: target mel [B, T, mel]; decoder input is the
        reduction-factor-subsampled teacher-forced mel."""
        enc = self.speecht5.encode(
            self.speecht5.text_encoder_prenet(input_ids))
        r = self.config.reduction_factor
        if decoder_mel is None:
            assert labels is not None
            # shift-right + keep every r-th frame (reduction)
            z = torch.zeros_like(labels[:, :1])
            decoder_mel = torch.cat([z, labels[:, :-1]], dim=1)[:, ::r]
        dec_in = self.speecht5.speech_decoder_prenet(decoder_mel)
        h = self.speecht5.decode(dec_in, enc)
        before, after, stop = self.speech_decoder_postnet(h)
        if labels is not None:
            T = min(before.shape[1], labels.shape[1])
            loss = F.l1_loss(before[:, :T], labels[:, :T]) + \
                F.l1_loss(after[:, :T], labels[:, :T])
            if stop_labels is not None:
                Ts = min(stop.shape[1], stop_labels.shape[1])
                loss = loss + F.binary_cross_entropy_with_logits(
                    stop[:, :Ts], stop_labels[:, :Ts].to(stop.dtype))
            return loss, after
        return before, after, stop
