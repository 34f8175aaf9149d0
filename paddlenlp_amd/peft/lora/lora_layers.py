"""LoRA layers, including tensor-parallel variants.

Reference behavior: paddlenlp/peft/lora/lora_layers.py (LoRALinear,
ColumnParallelLoRALinear/RowParallelLoRALinear with explicit collectives
:257-301), lora_config.py (rslora, lora_plus_scale).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...parallel.tensor_parallel import (
    ColumnParallelLinear,
    RowParallelLinear,
    copy_to_model_parallel,
    reduce_from_model_parallel,
)


class LoRALinear(nn.Module):
    """y = x W^T + scale * (dropout(x) A^T) B^T; only A/B train.

    `base` may be an nn.Linear or a QuantizationLinear (QLoRA: frozen 4-bit
    base + trainable adapters, reference peft/lora over quantization_linear)."""

    def __init__(self, base: nn.Module, r: int, lora_alpha: float = 1.0,
                 lora_dropout: float = 0.0, rslora: bool = False,
                 lora_plus_scale: float = 1.0):
        super().__init__()
        self.base = base
        self.r = r
        self.scaling = lora_alpha / math.sqrt(r) if rslora else lora_alpha / r
        self.lora_dropout = nn.Dropout(lora_dropout) if lora_dropout > 0 else nn.Identity()
        in_f = base.in_features
        out_f = base.out_features
        base_w = getattr(base, "weight", None)
        if base_w is not None:
            dtype, device = base_w.dtype, base_w.device
        else:  # quantized base: packed buffers, compute dtype recorded on it
            dtype = getattr(base, "compute_dtype", torch.float32)
            device = base.quant_weight.device
        self.lora_A = nn.Parameter(torch.zeros(r, in_f, dtype=dtype, device=device))
        self.lora_B = nn.Parameter(torch.zeros(out_f, r, dtype=dtype, device=device))
        # lora+ : B gets a higher LR via a param attribute the optimizer reads
        self.lora_B.lr_scale = lora_plus_scale
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        if base_w is not None:
            base_w.requires_grad_(False)
        if getattr(base, "bias", None) is not None:
            base.bias.requires_grad_(False)
        self.merged = False

    @property
    def weight(self):
        return getattr(self.base, "weight", None)

    def forward(self, x):
        y = self.base(x)
        if not self.merged:
            y = y + self.scaling * F.linear(F.linear(self.lora_dropout(x), self.lora_A), self.lora_B)
        return y

    @torch.no_grad()
    def merge(self):
        if self.merged:
            return
        if getattr(self.base, "weight", None) is None:
            raise RuntimeError("cannot merge LoRA into a quantized (packed) base")
        self.base.weight += self.scaling * (self.lora_B @ self.lora_A).to(self.base.weight.dtype)
        self.merged = True

    @torch.no_grad()
    def unmerge(self):
        if self.merged:
            self.base.weight -= self.scaling * (self.lora_B @ self.lora_A).to(self.base.weight.dtype)
            self.merged = False


class ColumnParallelLoRALinear(nn.Module):
    """LoRA on a ColumnParallelLinear: A replicated, B column-sharded
    (reference lora_layers.py:257-301)."""

    def __init__(self, base: ColumnParallelLinear, r: int, lora_alpha: float = 1.0,
                 lora_dropout: float = 0.0, rslora: bool = False,
                 lora_plus_scale: float = 1.0):
        super().__init__()
        self.base = base
        self.r = r
        self.scaling = lora_alpha / math.sqrt(r) if rslora else lora_alpha / r
        self.lora_dropout = nn.Dropout(lora_dropout) if lora_dropout > 0 else nn.Identity()
        dtype = base.weight.dtype
        device = base.weight.device
        self.lora_A = nn.Parameter(torch.zeros(r, base.in_features, dtype=dtype, device=device))
        self.lora_B = nn.Parameter(
            torch.zeros(base.out_features_per_partition, r, dtype=dtype, device=device))
        self.lora_B.is_column_parallel = True
        self.lora_B.lr_scale = lora_plus_scale
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        base.weight.requires_grad_(False)
        self.merged = False

    def forward(self, x):
        y = self.base(x)
        if not self.merged:
            xa = copy_to_model_parallel(self.lora_dropout(x), self.base.group)
            y = y + self.scaling * F.linear(F.linear(xa, self.lora_A), self.lora_B)
        return y

    @torch.no_grad()
    def merge(self):
        if not self.merged:
            self.base.weight += self.scaling * (self.lora_B @ self.lora_A).to(self.base.weight.dtype)
            self.merged = True


class RowParallelLoRALinear(nn.Module):
    """LoRA on a RowParallelLinear: A row-sharded, B replicated; the LoRA
    branch needs its own all-reduce like the base row linear."""

    def __init__(self, base: RowParallelLinear, r: int, lora_alpha: float = 1.0,
                 lora_dropout: float = 0.0, rslora: bool = False,
                 lora_plus_scale: float = 1.0):
        super().__init__()
        self.base = base
        self.r = r
        self.scaling = lora_alpha / math.sqrt(r) if rslora else lora_alpha / r
        self.lora_dropout = nn.Dropout(lora_dropout) if lora_dropout > 0 else nn.Identity()
        dtype = base.weight.dtype
        device = base.weight.device
        self.lora_A = nn.Parameter(
            torch.zeros(r, base.in_features_per_partition, dtype=dtype, device=device))
        self.lora_A.is_row_parallel = True
        self.lora_B = nn.Parameter(torch.zeros(base.out_features, r, dtype=dtype, device=device))
        self.lora_B.lr_scale = lora_plus_scale
        nn.init.kaiming_uniform_(self.lora_A, a=math.sqrt(5))
        base.weight.requires_grad_(False)
        self.merged = False

    def forward(self, x):
        y = self.base(x)
        if not self.merged:
            partial = F.linear(self.lora_dropout(x), self.lora_A)
            partial = reduce_from_model_parallel(partial, self.base.group)
            y = y + self.scaling * F.linear(partial, self.lora_B)
        return y

    @torch.no_grad()
    def merge(self):
        if not self.merged:
            self.base.weight += self.scaling * (self.lora_B @ self.lora_A).to(self.base.weight.dtype)
            self.merged = True
