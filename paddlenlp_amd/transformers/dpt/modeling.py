"""DPT — Dense Prediction Transformer (reference:
paddlenlp/transformers/dpt/modeling.py).

ViT backbone whose intermediate layers are REASSEMBLED into multi-scale
feature maps (the DPT reassemble stage: readout-token fusion + per-level
resize convs), fused top-down with residual conv units, and decoded by a
depth-estimation (or semantic-segmentation) head.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..clip.modeling import _PreLNLayer
from ..model_utils import PretrainedModel

__all__ = ["DPTConfig", "DPTModel", "DPTForDepthEstimation"]


class DPTConfig(PretrainedConfig):
    model_type = "dpt"

    def __init__(self, hidden_size=768, num_hidden_layers=12,
                 num_attention_heads=12, intermediate_size=3072,
                 image_size=384, patch_size=16, num_channels=3,
                 backbone_out_indices=(2, 5, 8, 11),
                 neck_hidden_sizes=(96, 192, 384, 768),
                 fusion_hidden_size=256, layer_norm_eps=1e-12,
                 initializer_range=0.02, **kwargs):
        super().__init__(**kwargs)
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.image_size = image_size
        self.patch_size = patch_size
        self.num_channels = num_channels
        self.backbone_out_indices = list(backbone_out_indices)
        self.neck_hidden_sizes = list(neck_hidden_sizes)
        self.fusion_hidden_size = fusion_hidden_size
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range


class DPTPretrainedModel(PretrainedModel):
    config_class = DPTConfig
    base_model_prefix = "dpt"

    def _init_weights(self, module):
        if isinstance(module, (nn.Linear, nn.Conv2d)):
            module.weight.data.normal_(std=self.config.initializer_range)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.LayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()


class DPTModel(DPTPretrainedModel):
    """ViT returning the hidden states at backbone_out_indices."""

    def __init__(self, config: DPTConfig):
        super().__init__(config)
        c = config
        self.patch_embed = nn.Conv2d(c.num_channels, c.hidden_size,
                                     kernel_size=c.patch_size,
                                     stride=c.patch_size)
        n = (c.image_size // c.patch_size) ** 2
        self.cls_token = nn.Parameter(torch.zeros(1, 1, c.hidden_size))
        self.pos_embed = nn.Parameter(torch.zeros(1, n + 1, c.hidden_size))
        self.layers = nn.ModuleList(
            [_PreLNLayer(c.hidden_size, c.num_attention_heads,
                         c.intermediate_size, c.layer_norm_eps)
             for _ in range(c.num_hidden_layers)])
        self.init_weights()

    def forward(self, pixel_values):
        B = pixel_values.shape[0]
        x = self.patch_embed(pixel_values).flatten(2).transpose(1, 2)
        x = torch.cat([self.cls_token.expand(B, 1, -1), x], dim=1)
        x = x + self.pos_embed[:, :x.shape[1]]
        feats = []
        for i, layer in enumerate(self.layers):
            x = layer(x)
            if i in self.config.backbone_out_indices:
                feats.append(x)
        return x, feats


class _ResidualConvUnit(nn.Module):
    def __init__(self, ch):
        super().__init__()
        self.conv1 = nn.Conv2d(ch, ch, 3, padding=1)
        self.conv2 = nn.Conv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        h = self.conv1(F.relu(x))
        return x + self.conv2(F.relu(h))


class DPTForDepthEstimation(DPTPretrainedModel):
    def __init__(self, config: DPTConfig):
        super().__init__(config)
        self.dpt = DPTModel(config)
        c = config
        # reassemble: per-level projection (readout token added back) +
        # resize to the level's scale (4x, 2x, 1x, 0.5x of patch grid)
        self.reassemble_proj = nn.ModuleList(
            [nn.Conv2d(c.hidden_size, ns, 1) for ns in c.neck_hidden_sizes])
        self.readout_proj = nn.ModuleList(
            [nn.Linear(2 * c.hidden_size, c.hidden_size)
             for _ in c.neck_hidden_sizes])
        self.neck_convs = nn.ModuleList(
            [nn.Conv2d(ns, c.fusion_hidden_size, 3, padding=1, bias=False)
             for ns in c.neck_hidden_sizes])
        self.fusion_units = nn.ModuleList(
            [_ResidualConvUnit(c.fusion_hidden_size)
             for _ in c.neck_hidden_sizes])
        f = c.fusion_hidden_size
        self.head = nn.Sequential(
            nn.Conv2d(f, f // 2, 3, padding=1), nn.ReLU(),
            nn.Conv2d(f // 2, 32, 3, padding=1), nn.ReLU(),
            nn.Conv2d(32, 1, 1))
        self.init_weights()

    def forward(self, pixel_values, labels=None):
        B = pixel_values.shape[0]
        g = pixel_values.shape[-1] // self.config.patch_size
        _, feats = self.dpt(pixel_values)
        scales = [4.0, 2.0, 1.0, 0.5]
        levels = []
        for i, f in enumerate(feats):
            cls, patches = f[:, :1], f[:, 1:]
            # readout fusion: concat cls onto every patch (reference
            # readout_type="project")
            p = self.readout_proj[i](
                torch.cat([patches, cls.expand_as(patches)], dim=-1))
            fm = p.transpose(1, 2).reshape(B, -1, g, g)
            fm = self.reassemble_proj[i](fm)
            fm = F.interpolate(fm, scale_factor=scales[i], mode="bilinear",
                               align_corners=False)
            levels.append(self.neck_convs[i](fm))
        # top-down fusion: coarsest first, upsample + residual units
        x = self.fusion_units[-1](levels[-1])
        for i in range(len(levels) - 2, -1, -1):
            x = F.interpolate(x, size=levels[i].shape[-2:], mode="bilinear",
                              align_corners=False)
            x = self.fusion_units[i](x + levels[i])
        depth = self.head(x).squeeze(1)
        if labels is not None:
            size = labels.shape[-2:]
            pred = F.interpolate(depth[:, None], size=size,
                                 mode="bilinear",
                                 align_corners=False).squeeze(1)
            return F.l1_loss(pred, labels), depth
        return depth
