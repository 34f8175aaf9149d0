"""BiT — Big Transfer ResNet v2 (reference:
paddlenlp/transformers/bit/modeling.py).

Pre-activation ResNet with the two BiT signatures: GroupNorm everywhere
(no BatchNorm) and WEIGHT-STANDARDIZED convolutions (conv weights
normalized to zero mean / unit variance per output channel at every
forward).  Used as the visual backbone for DPT-style heads.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..model_utils import PretrainedModel

__all__ = ["BitConfig", "BitModel", "BitForImageClassification"]


class BitConfig(PretrainedConfig):
    model_type = "bit"

    def __init__(self, num_channels=3, embedding_size=64,
                 hidden_sizes=(256, 512, 1024, 2048),
                 depths=(3, 4, 6, 3), num_groups=32,
                 initializer_range=0.02, num_labels=1000, **kwargs):
        super().__init__(**kwargs)
        self.num_channels = num_channels
        self.embedding_size = embedding_size
        self.hidden_sizes = list(hidden_sizes)
        self.depths = list(depths)
        self.num_groups = num_groups
        self.initializer_range = initializer_range
        self.num_labels = num_labels


class WSConv2d(nn.Conv2d):
    """Weight-standardized conv (the BiT trick)."""

    def forward(self, x):
        w = self.weight
        mean = w.mean(dim=(1, 2, 3), keepdim=True)
        var = w.var(dim=(1, 2, 3), keepdim=True, unbiased=False)
        w = (w - mean) * torch.rsqrt(var + 1e-10)
        return F.conv2d(x, w, self.bias, self.stride, self.padding,
                        self.dilation, self.groups)


def _gn(groups, ch):
    return nn.GroupNorm(min(groups, ch), ch)


class BitBottleneck(nn.Module):
    """Pre-activation bottleneck: GN-ReLU-conv x3 + shortcut."""

    def __init__(self, cin, cout, stride, groups):
        super().__init__()
        mid = cout // 4
        self.norm1 = _gn(groups, cin)
        self.conv1 = WSConv2d(cin, mid, 1, bias=False)
        self.norm2 = _gn(groups, mid)
        self.conv2 = WSConv2d(mid, mid, 3, stride=stride, padding=1,
                              bias=False)
        self.norm3 = _gn(groups, mid)
        self.conv3 = WSConv2d(mid, cout, 1, bias=False)
        self.shortcut = None
        if cin != cout or stride != 1:
            self.shortcut = WSConv2d(cin, cout, 1, stride=stride,
                                     bias=False)

    def forward(self, x):
        h = F.relu(self.norm1(x))
        sc = self.shortcut(h) if self.shortcut is not None else x
        h = self.conv1(h)
        h = self.conv2(F.relu(self.norm2(h)))
        h = self.conv3(F.relu(self.norm3(h)))
        return h + sc


class BitPretrainedModel(PretrainedModel):
    config_class = BitConfig
    base_model_prefix = "bit"

    def _init_weights(self, module):
        if isinstance(module, nn.Conv2d):
            nn.init.kaiming_normal_(module.weight, mode="fan_out",
                                    nonlinearity="relu")
        elif isinstance(module, nn.Linear):
            module.weight.data.normal_(std=self.config.initializer_range)
            if module.bias is not None:
                module.bias.data.zero_()


class BitModel(BitPretrainedModel):
    def __init__(self, config: BitConfig):
        super().__init__(config)
        g = config.num_groups
        self.embedder = WSConv2d(config.num_channels,
                                 config.embedding_size, 7, stride=2,
                                 padding=3, bias=False)
        stages = []
        cin = config.embedding_size
        for si, (cout, depth) in enumerate(zip(config.hidden_sizes,
                                               config.depths)):
            blocks = []
            for bi in range(depth):
                stride = 2 if (bi == 0 and si > 0) else 1
                blocks.append(BitBottleneck(cin, cout, stride, g))
                cin = cout
            stages.append(nn.Sequential(*blocks))
        self.stages = nn.ModuleList(stages)
        self.final_norm = _gn(g, cin)
        self.init_weights()

    def forward(self, pixel_values, output_hidden_states=False):
        x = self.embedder(pixel_values)
        x = F.max_pool2d(x, 3, stride=2, padding=1)
        hidden = []
        for stage in self.stages:
            x = stage(x)
            hidden.append(x)
        x = F.relu(self.final_norm(x))
        pooled = x.mean(dim=(2, 3))
        if output_hidden_states:
            return x, pooled, hidden
        return x, pooled


class BitForImageClassification(BitPretrainedModel):
    def __init__(self, config: BitConfig):
        super().__init__(config)
        self.bit = BitModel(config)
        self.classifier = nn.Linear(config.hidden_sizes[-1],
                                    config.num_labels)
        self.init_weights()

    def forward(self, pixel_values, labels=None):
        _, pooled = self.bit(pixel_values)
        logits = self.classifier(pooled)
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
