"""LR scheduler classes with the reference's names.

Reference: paddlenlp/transformers/optimization.py —
CosineAnnealingWithWarmupDecay :33, LinearAnnealingWithWarmupDecay :55
(used by llm/run_pretrain.py:522-536).
"""
from __future__ import annotations

import math

import torch


class CosineAnnealingWithWarmupDecay(torch.optim.lr_scheduler.LambdaLR):
    def __init__(self, optimizer, max_lr, min_lr, warmup_step, decay_step, last_epoch=-1):
        ratio = min_lr / max_lr if max_lr else 0.0

        def fn(step):
            if warmup_step > 0 and step < warmup_step:
                return step / warmup_step
            if step >= decay_step:
                return ratio
            progress = (step - warmup_step) / max(1, decay_step - warmup_step)
            return ratio + (1 - ratio) * 0.5 * (1 + math.cos(math.pi * progress))

        super().__init__(optimizer, fn, last_epoch=last_epoch)


class LinearAnnealingWithWarmupDecay(torch.optim.lr_scheduler.LambdaLR):
    def __init__(self, optimizer, max_lr, min_lr, warmup_step, decay_step, last_epoch=-1):
        ratio = min_lr / max_lr if max_lr else 0.0

        def fn(step):
            if warmup_step > 0 and step < warmup_step:
                return step / warmup_step
            if step >= decay_step:
                return ratio
            progress = (step - warmup_step) / max(1, decay_step - warmup_step)
            return ratio + (1 - ratio) * (1 - progress)

        super().__init__(optimizer, fn, last_epoch=last_epoch)
