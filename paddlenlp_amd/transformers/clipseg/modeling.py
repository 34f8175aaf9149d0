"""CLIPSeg (reference: paddlenlp/transformers/clipseg/modeling.py).

Zero-shot segmentation over a frozen CLIP: intermediate ViT activations
(extract_layers) are reduced and run through a small decoder whose
blocks are FiLM-conditioned on the text (or visual-prompt) embedding —
conditional scale+shift per block (reference film_mul/film_add) — and a
transposed-conv head upsamples to a dense mask.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..clip.configuration import CLIPTextConfig, CLIPVisionConfig
from ..clip.modeling import CLIPTextTransformer, CLIPVisionTransformer
from ..configuration_utils import PretrainedConfig
from ..encoder import init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["CLIPSegConfig", "CLIPSegForImageSegmentation"]


class CLIPSegConfig(PretrainedConfig):
    model_type = "clipseg"

    def __init__(self, text_config=None, vision_config=None,
                 projection_dim=512, extract_layers=(3, 6, 9),
                 reduce_dim=64, decoder_num_attention_heads=4,
                 decoder_intermediate_size=2048, **kwargs):
        super().__init__(**kwargs)
        self.text_config = CLIPTextConfig(**(text_config or {}))
        self.vision_config = CLIPVisionConfig(**(vision_config or {}))
        self.projection_dim = projection_dim
        self.extract_layers = list(extract_layers)
        self.reduce_dim = reduce_dim
        self.decoder_num_attention_heads = decoder_num_attention_heads
        self.decoder_intermediate_size = decoder_intermediate_size
        self.initializer_range = 0.02


class _DecoderBlock(nn.Module):
    def __init__(self, config: CLIPSegConfig):
        super().__init__()
        d = config.reduce_dim
        self.attn = nn.MultiheadAttention(
            d, config.decoder_num_attention_heads, batch_first=True)
        self.norm1 = nn.LayerNorm(d)
        self.fc1 = nn.Linear(d, config.decoder_intermediate_size)
        self.fc2 = nn.Linear(config.decoder_intermediate_size, d)
        self.norm2 = nn.LayerNorm(d)

    def forward(self, x):
        a, _ = self.attn(x, x, x, need_weights=False)
        x = self.norm1(x + a)
        return self.norm2(x + self.fc2(F.relu(self.fc1(x))))


class CLIPSegForImageSegmentation(PretrainedModel):
    config_class = CLIPSegConfig
    base_model_prefix = "clipseg"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)

    def __init__(self, config: CLIPSegConfig):
        super().__init__(config)
        self.vision_model = CLIPVisionTransformer(config.vision_config)
        self.text_model = CLIPTextTransformer(config.text_config)
        self.text_projection = nn.Linear(config.text_config.hidden_size,
                                         config.projection_dim, bias=False)
        d = config.reduce_dim
        n = len(config.extract_layers)
        self.reduces = nn.ModuleList(
            [nn.Linear(config.vision_config.hidden_size, d)
             for _ in range(n)])
        self.blocks = nn.ModuleList(
            [_DecoderBlock(config) for _ in range(n)])
        # FiLM conditioning from the text embedding (the CLIPSeg trick)
        self.film_mul = nn.Linear(config.projection_dim, d)
        self.film_add = nn.Linear(config.projection_dim, d)
        p = config.vision_config.patch_size
        self.head = nn.Sequential(
            nn.Conv2d(d, d, 3, padding=1), nn.ReLU(),
            nn.ConvTranspose2d(d, d // 2, p // 2, stride=p // 2), nn.ReLU(),
            nn.ConvTranspose2d(d // 2, 1, 2, stride=2))
        self.init_weights()

    def _vision_activations(self, pixel_values):
        vm = self.vision_model
        B = pixel_values.shape[0]
        patches = vm.patch_embedding(pixel_values).flatten(2).transpose(1, 2)
        cls = vm.class_embedding[None, None].expand(B, 1, -1)
        x = torch.cat([cls, patches], dim=1)
        pos = torch.arange(x.shape[1], device=x.device)
        x = vm.pre_layernorm(x + vm.position_embedding(pos))
        acts = []
        for i, layer in enumerate(vm.layers):
            x = layer(x)
            if i in self.config.extract_layers:
                acts.append(x)
        return acts

    def forward(self, input_ids, pixel_values, labels=None):
        cond = self.text_projection(self.text_model(input_ids)[1])
        acts = self._vision_activations(pixel_values)
        B = pixel_values.shape[0]
        g = pixel_values.shape[-1] // self.config.vision_config.patch_size
        x = None
        # deepest activation first, FiLM applied at the first block
        for i, (act, red, blk) in enumerate(
                zip(reversed(acts), self.reduces, self.blocks)):
            a = red(act[:, 1:])            # drop class token
            x = a if x is None else x + a
            if i == 0:
                x = self.film_mul(cond)[:, None] * x + \
                    self.film_add(cond)[:, None]
            x = blk(x)
        fm = x.transpose(1, 2).reshape(B, -1, g, g)
        logits = self.head(fm).squeeze(1)
        if labels is not None:
            pred = F.interpolate(logits[:, None], size=labels.shape[-2:],
                                 mode="bilinear",
                                 align_corners=False).squeeze(1)
            return F.binary_cross_entropy_with_logits(
                pred, labels.to(pred.dtype)), logits
        return logits
