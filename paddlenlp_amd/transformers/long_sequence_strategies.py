"""Pluggable long-sequence strategies.

Reference behavior: paddlenlp/transformers/long_sequence_strategies/
(long_sequence_strategies.py factory + embedding_strategies.py RotaryEmbedding
/LinearScaling/NTKScaling/YaRN + attention_strategies.py
AttentionWithLinearBias).
"""
from __future__ import annotations

import math

import torch


class RotaryEmbedding:
    def __init__(self, head_dim: int, base: float = 10000.0, scaling_factor: float = 1.0):
        self.head_dim = head_dim
        self.base = base
        self.scaling_factor = scaling_factor

    def inv_freq(self, device=None):
        d = self.head_dim
        return 1.0 / (self.base ** (torch.arange(0, d, 2, dtype=torch.float32, device=device) / d))

    def position_scale(self):
        return 1.0

    def tables(self, seq_len: int, device=None):
        t = torch.arange(seq_len, dtype=torch.float32, device=device) * self.position_scale()
        freqs = torch.outer(t, self.inv_freq(device))
        emb = torch.cat((freqs, freqs), dim=-1)
        return emb.cos(), emb.sin()


class LinearScalingRotaryEmbedding(RotaryEmbedding):
    def position_scale(self):
        return 1.0 / self.scaling_factor


class NTKScalingRotaryEmbedding(RotaryEmbedding):
    def inv_freq(self, device=None):
        d = self.head_dim
        base = self.base * self.scaling_factor ** (d / (d - 2))
        return 1.0 / (base ** (torch.arange(0, d, 2, dtype=torch.float32, device=device) / d))


class YaRNScalingRotaryEmbedding(RotaryEmbedding):
    """NTK-by-parts interpolation (YaRN) with the standard beta window."""

    def __init__(self, head_dim, base=10000.0, scaling_factor=1.0,
                 original_max_position=2048, beta_fast=32.0, beta_slow=1.0):
        super().__init__(head_dim, base, scaling_factor)
        self.orig = original_max_position
        self.beta_fast = beta_fast
        self.beta_slow = beta_slow

    def _correction_dim(self, num_rotations):
        return (self.head_dim * math.log(self.orig / (num_rotations * 2 * math.pi))
                / (2 * math.log(self.base)))

    def inv_freq(self, device=None):
        base_inv = super().inv_freq(device)
        interp = base_inv / self.scaling_factor
        low = math.floor(self._correction_dim(self.beta_fast))
        high = math.ceil(self._correction_dim(self.beta_slow))
        low, high = max(low, 0), min(high, self.head_dim // 2 - 1)
        ramp = torch.clamp(
            (torch.arange(self.head_dim // 2, dtype=torch.float32, device=device) - low)
            / max(1e-3, high - low), 0, 1)
        return base_inv * (1 - ramp) + interp * ramp

    def attention_factor(self):
        return 0.1 * math.log(self.scaling_factor) + 1.0


class AttentionWithLinearBias:
    """ALiBi slopes (reference attention_strategies.py)."""

    @staticmethod
    def slopes(num_heads: int) -> torch.Tensor:
        closest = 2 ** math.floor(math.log2(num_heads))
        base = torch.tensor([2 ** (-8.0 * (i + 1) / closest) for i in range(closest)])
        if closest < num_heads:
            extra = torch.tensor(
                [2 ** (-4.0 * (i + 1) / closest) for i in range(num_heads - closest)])
            base = torch.cat([base, extra])
        return base

    @staticmethod
    def bias(num_heads: int, seq_len: int, device=None) -> torch.Tensor:
        """[H, 1, S] additive bias."""
        slopes = AttentionWithLinearBias.slopes(num_heads).to(device)
        pos = torch.arange(seq_len, device=device).float()
        return -slopes[:, None, None] * pos[None, None, :]


STRATEGIES = {
    ("embedding", "RotaryEmbedding"): RotaryEmbedding,
    ("embedding", "LinearScalingRotaryEmbedding"): LinearScalingRotaryEmbedding,
    ("embedding", "NTKScalingRotaryEmbedding"): NTKScalingRotaryEmbedding,
    ("embedding", "YaRNScalingRotaryEmbedding"): YaRNScalingRotaryEmbedding,
    ("attention", "AttentionWithLinearBias"): AttentionWithLinearBias,
}


class LongSequenceStrategies:
    @classmethod
    def build_long_sequence_strategy(cls, strategy_type: str, strategy_name: str, **kwargs):
        key = (strategy_type, strategy_name)
        if key not in STRATEGIES:
            raise ValueError(f"Unknown long-sequence strategy {key}; have {sorted(STRATEGIES)}")
        klass = STRATEGIES[key]
        return klass(**kwargs) if strategy_type == "embedding" else klass
