"""DALL-E mini / DALLE-BART (reference:
paddlenlp/transformers/dallebart/modeling.py).

Text-to-image as seq2seq over VQ tokens: a BART-style text encoder and
an image-token decoder with SEPARATE vocabularies (text_vocab_size vs
image_vocab_size + BOS), GLU feed-forward blocks (the dallebart delta
vs vanilla BART), and sampling of image token grids for the VQGAN
detokenizer (the VQGAN itself is an external module in the reference
too).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...generation import GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["DalleBartConfig", "DalleBartModel",
           "DalleBartForConditionalGeneration"]


class DalleBartConfig(PretrainedConfig):
    model_type = "dallebart"

    def __init__(self, text_vocab_size=50264, image_vocab_size=16384,
                 hidden_size=1024, num_encoder_layers=12,
                 num_decoder_layers=12, num_attention_heads=16,
                 intermediate_size=2730, activation_function="gelu",
                 dropout=0.0, max_text_length=64, image_length=256,
                 initializer_range=0.02, layer_norm_eps=1e-5,
                 pad_token_id=1, bos_token_id=16384, **kwargs):
        super().__init__(**kwargs)
        self.text_vocab_size = text_vocab_size
        self.image_vocab_size = image_vocab_size
        self.vocab_size = image_vocab_size + 1       # + image BOS
        self.hidden_size = hidden_size
        self.num_encoder_layers = num_encoder_layers
        self.num_decoder_layers = num_decoder_layers
        self.num_hidden_layers = num_encoder_layers + num_decoder_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.activation_function = activation_function
        self.dropout = dropout
        self.max_text_length = max_text_length
        self.image_length = image_length
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class GLUFeedForward(nn.Module):
    """ln -> (gelu(W1 x) * V x) -> W2: the dallebart GLU block."""

    def __init__(self, config):
        super().__init__()
        h, m = config.hidden_size, config.intermediate_size
        self.ln = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.w1 = nn.Linear(h, m)
        self.v = nn.Linear(h, m)
        self.w2 = nn.Linear(m, h)

    def forward(self, x):
        x = self.ln(x)
        return self.w2(F.gelu(self.w1(x)) * self.v(x))


class _Layer(nn.Module):
    def __init__(self, config, cross=False):
        super().__init__()
        h = config.hidden_size
        self.nh, self.dh = config.num_attention_heads, config.head_dim
        self.attn_ln = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.qkv = nn.Linear(h, 3 * h)
        self.out = nn.Linear(h, h)
        self.cross = cross
        if cross:
            self.cross_ln = nn.LayerNorm(h, eps=config.layer_norm_eps)
            self.cq = nn.Linear(h, h)
            self.ckv = nn.Linear(h, 2 * h)
            self.cout = nn.Linear(h, h)
        self.glu = GLUFeedForward(config)

    def _sa(self, x, causal):
        B, S, H = x.shape
        q, k, v = self.qkv(x).chunk(3, dim=-1)
        shp = (B, S, self.nh, self.dh)
        o = F.scaled_dot_product_attention(
            q.view(shp).transpose(1, 2), k.view(shp).transpose(1, 2),
            v.view(shp).transpose(1, 2), is_causal=causal)
        return self.out(o.transpose(1, 2).reshape(B, S, H))

    def forward(self, x, enc=None, causal=False):
        x = x + self._sa(self.attn_ln(x), causal)
        if self.cross and enc is not None:
            B, S, H = x.shape
            Se = enc.shape[1]
            h = self.cross_ln(x)
            q = self.cq(h).view(B, S, self.nh, self.dh).transpose(1, 2)
            k, v = self.ckv(enc).chunk(2, dim=-1)
            k = k.view(B, Se, self.nh, self.dh).transpose(1, 2)
            v = v.view(B, Se, self.nh, self.dh).transpose(1, 2)
            o = F.scaled_dot_product_attention(q, k, v)
            x = x + self.cout(o.transpose(1, 2).reshape(B, S, H))
        return x + self.glu(x)


class DalleBartPretrainedModel(PretrainedModel):
    config_class = DalleBartConfig
    base_model_prefix = "dallebart"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class DalleBartModel(DalleBartPretrainedModel):
    def __init__(self, config: DalleBartConfig):
        super().__init__(config)
        h = config.hidden_size
        # separate text / image vocabularies (the dallebart signature)
        self.text_embed = nn.Embedding(config.text_vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.text_pos = nn.Embedding(config.max_text_length, h)
        self.image_embed = nn.Embedding(config.image_vocab_size + 1, h)
        self.image_pos = nn.Embedding(config.image_length, h)
        self.encoder = nn.ModuleList(
            [_Layer(config) for _ in range(config.num_encoder_layers)])
        self.decoder = nn.ModuleList(
            [_Layer(config, cross=True)
             for _ in range(config.num_decoder_layers)])
        self.enc_ln = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.dec_ln = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.init_weights()

    def get_input_embeddings(self):
        return self.text_embed

    def encode(self, input_ids):
        pos = torch.arange(input_ids.shape[1], device=input_ids.device)
        x = self.text_embed(input_ids) + self.text_pos(pos)
        for layer in self.encoder:
            x = layer(x)
        return self.enc_ln(x)

    def decode(self, image_ids, enc):
        pos = torch.arange(image_ids.shape[1], device=image_ids.device)
        x = self.image_embed(image_ids) + self.image_pos(pos)
        for layer in self.decoder:
            x = layer(x, enc, causal=True)
        return self.dec_ln(x)

    def forward(self, input_ids, decoder_input_ids):
        return self.decode(decoder_input_ids, self.encode(input_ids))


class DalleBartForConditionalGeneration(DalleBartPretrainedModel,
                                        GenerationMixin):
    def __init__(self, config: DalleBartConfig):
        super().__init__(config)
        self.dallebart = DalleBartModel(config)
        self.lm_head = nn.Linear(config.hidden_size,
                                 config.image_vocab_size + 1, bias=False)

    def forward(self, input_ids, decoder_input_ids=None, labels=None,
                **kwargs):
        if decoder_input_ids is None and labels is not None:
            bos = torch.full((labels.shape[0], 1), self.config.bos_token_id,
                             dtype=labels.dtype, device=labels.device)
            decoder_input_ids = torch.cat([bos, labels[:, :-1].clamp(min=0)],
                                          dim=1)
        logits = self.lm_head(self.dallebart(input_ids, decoder_input_ids))
        if labels is not None:
            loss = F.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                ignore_index=-100)
            return loss, logits
        return logits

    @torch.no_grad()
    def generate_image_tokens(self, input_ids, temperature=1.0, top_k=0):
        """Sample an image_length token grid conditioned on the text."""
        enc = self.dallebart.encode(input_ids)
        B = input_ids.shape[0]
        ids = torch.full((B, 1), self.config.bos_token_id,
                         dtype=torch.long, device=input_ids.device)
        for _ in range(self.config.image_length):
            logits = self.lm_head(self.dallebart.decode(ids, enc))[:, -1]
            logits = logits / max(temperature, 1e-5)
            if top_k:
                kth = logits.topk(top_k, dim=-1).values[:, -1:]
                logits = logits.masked_fill(logits < kth, -float("inf"))
            probs = logits.softmax(-1)
            ids = torch.cat([ids, torch.multinomial(probs, 1)], dim=1)
        return ids[:, 1:]
