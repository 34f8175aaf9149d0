"""BLIP-2 (reference: paddlenlp/transformers/blip_2/modeling.py).

Frozen-backbone VLM bridge: a ViT vision encoder, the Q-FORMER — a set
of learned query tokens run through self-attention + cross-attention
into the image patches (reference Blip2QFormerMultiHeadAttention :487,
query_tokens) — and a language-model projection that hands the 32 query
outputs to any causal LM as soft visual prompts
(Blip2ForConditionalGeneration: vision -> qformer -> projection -> LM).
The LM is pluggable (any framework CausalLM); BLIP-2's contribution is
the bridge, which is what lives here.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..clip.configuration import CLIPVisionConfig
from ..clip.modeling import CLIPVisionTransformer
from ..encoder import ACT2FN, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["Blip2Config", "Blip2QFormerModel", "Blip2Model"]


class Blip2QFormerConfig(PretrainedConfig):
    model_type = "blip_2_qformer"

    def __init__(self, hidden_size=768, num_hidden_layers=12,
                 num_attention_heads=12, intermediate_size=3072,
                 cross_attention_frequency=2, encoder_hidden_size=1408,
                 hidden_act="gelu", layer_norm_eps=1e-12,
                 initializer_range=0.02, **kwargs):
        super().__init__(**kwargs)
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.cross_attention_frequency = cross_attention_frequency
        self.encoder_hidden_size = encoder_hidden_size
        self.hidden_act = hidden_act
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class Blip2Config(PretrainedConfig):
    model_type = "blip-2"

    def __init__(self, vision_config=None, qformer_config=None,
                 num_query_tokens=32, lm_hidden_size=2048, **kwargs):
        super().__init__(**kwargs)
        self.vision_config = CLIPVisionConfig(**(vision_config or {}))
        self.qformer_config = Blip2QFormerConfig(**(qformer_config or {}))
        self.qformer_config.encoder_hidden_size = \
            self.vision_config.hidden_size
        self.num_query_tokens = num_query_tokens
        self.lm_hidden_size = lm_hidden_size
        self.initializer_range = self.qformer_config.initializer_range


class _MHA(nn.Module):
    def __init__(self, config, kv_size=None):
        super().__init__()
        h = config.hidden_size
        self.nh, self.dh = config.num_attention_heads, config.head_dim
        self.query = nn.Linear(h, h)
        self.key = nn.Linear(kv_size or h, h)
        self.value = nn.Linear(kv_size or h, h)
        self.out = nn.Linear(h, h)

    def forward(self, q_in, kv_in):
        B, Sq, H = q_in.shape
        Sk = kv_in.shape[1]
        q = self.query(q_in).view(B, Sq, self.nh, self.dh).transpose(1, 2)
        k = self.key(kv_in).view(B, Sk, self.nh, self.dh).transpose(1, 2)
        v = self.value(kv_in).view(B, Sk, self.nh, self.dh).transpose(1, 2)
        o = F.scaled_dot_product_attention(q, k, v)
        return self.out(o.transpose(1, 2).reshape(B, Sq, H))


class Blip2QFormerLayer(nn.Module):
    """Self-attention over queries + (every Nth layer) cross-attention
    into the frozen image features (reference :636-700)."""

    def __init__(self, config: Blip2QFormerConfig, has_cross: bool):
        super().__init__()
        h = config.hidden_size
        self.self_attn = _MHA(config)
        self.self_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.cross_attn = (_MHA(config, config.encoder_hidden_size)
                           if has_cross else None)
        self.cross_norm = (nn.LayerNorm(h, eps=config.layer_norm_eps)
                           if has_cross else None)
        self.ff_in = nn.Linear(h, config.intermediate_size)
        self.ff_out = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, image_embeds):
        x = self.self_norm(x + self.self_attn(x, x))
        if self.cross_attn is not None:
            x = self.cross_norm(x + self.cross_attn(x, image_embeds))
        return self.ff_norm(x + self.ff_out(self.act(self.ff_in(x))))


class Blip2PretrainedModel(PretrainedModel):
    config_class = Blip2Config
    base_model_prefix = "blip2"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class Blip2QFormerModel(Blip2PretrainedModel):
    config_class = Blip2QFormerConfig

    def __init__(self, config: Blip2QFormerConfig):
        super().__init__(config)
        freq = config.cross_attention_frequency
        self.layers = nn.ModuleList([
            Blip2QFormerLayer(config, has_cross=(i % freq == 0))
            for i in range(config.num_hidden_layers)])
        self.norm = nn.LayerNorm(config.hidden_size,
                                 eps=config.layer_norm_eps)
        self.init_weights()

    def forward(self, query_embeds, image_embeds):
        x = query_embeds
        for layer in self.layers:
            x = layer(x, image_embeds)
        return self.norm(x)


class Blip2Model(Blip2PretrainedModel):
    """vision -> Q-Former -> language projection (the BLIP-2 bridge)."""

    def __init__(self, config: Blip2Config):
        super().__init__(config)
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        self.query_tokens = nn.Parameter(torch.zeros(
            1, config.num_query_tokens, config.qformer_config.hidden_size))
        nn.init.normal_(self.query_tokens, std=config.initializer_range)
        self.qformer = Blip2QFormerModel(config.qformer_config)
        self.language_projection = nn.Linear(
            config.qformer_config.hidden_size, config.lm_hidden_size)
        self.init_weights()

    def forward(self, pixel_values):
        """Returns the soft visual prompt [B, num_query_tokens,
        lm_hidden_size] to prepend to a causal LM's input embeddings."""
        image_embeds, _ = self.vision_model(pixel_values)
        q = self.query_tokens.expand(pixel_values.shape[0], -1, -1)
        return self.language_projection(self.qformer(q, image_embeds))

    def generate_inputs_for_lm(self, pixel_values, input_embeds):
        """Prepend the visual prompt to the LM's token embeddings."""
        prompt = self.forward(pixel_values)
        return torch.cat([prompt, input_embeds], dim=1)
