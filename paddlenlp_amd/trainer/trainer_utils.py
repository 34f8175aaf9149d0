"""Trainer utilities (reference: paddlenlp/trainer/trainer_utils.py).

set_seed, speed_metrics (:351 — tokens/s/device + hardware TFLOPS for MFU),
LR scheduler factory get_scheduler, ShardingOption (:315), checkpoint
discovery get_last_checkpoint, and the LLM FLOPs formula
(reference: paddlenlp/transformers/utils.py:963 caculate_llm_flops,
Megatron eq. 3).
"""
from __future__ import annotations

import math
import os
import random
import re
import time
from enum import Enum
from typing import Dict, Optional

import numpy as np
import torch

from ..utils.env import PREFIX_CHECKPOINT_DIR

# MI355X dense bf16 peak (MFU denominator): 2.5 PFLOP/s
# (AMD's 5 PF marketing figure includes 2:1 sparsity.)
MI355X_BF16_PEAK_FLOPS = 2.5e15
MI355X_FP8_PEAK_FLOPS = 5.0e15


class ShardingOption(Enum):
    """Reference: trainer_utils.py ShardingOption :315."""

    SHARD_OP = "stage1"
    SHARD_GRAD_OP = "stage2"
    FULL_SHARD = "stage3"
    OFFLOAD = "offload"


class IntervalStrategy(Enum):
    NO = "no"
    STEPS = "steps"
    EPOCH = "epoch"


def set_seed(seed: int):
    random.seed(seed)
    np.random.seed(seed)
    torch.manual_seed(seed)
    if torch.cuda.is_available():
        torch.cuda.manual_seed_all(seed)


def set_hybrid_seed(seed: int, topology=None):
    """Per-rank seed offsets: dp/sharding ranks must differ for dropout,
    mp/pp ranks must match for parameter init on replicated params."""
    from ..parallel.topology import get_topology

    topo = topology or get_topology()
    # global seed for replicated-state ops; offset by data-parallel coordinate
    set_seed(seed + topo.dataset_rank * 1024)


def caculate_llm_flops(
    hidden_size: int,
    intermediate_size: int,
    layer_num: int,
    vocab_size: int,
    seq_length: int,
    batch_size: int = 1,
    recompute: bool = False,
) -> int:
    """Megatron eq.3 FLOPs per optimization step for causal LM training.

    Reference: paddlenlp/transformers/utils.py:963-1003 (name kept with the
    reference's spelling for parity).
    """
    flops_per_token = (
        # attention qkvo + mlp
        2 * (4 * hidden_size * hidden_size + 3 * hidden_size * intermediate_size)
        # attention scores + context
        + 4 * hidden_size * seq_length
    )
    flops_fwd = batch_size * seq_length * layer_num * flops_per_token
    # lm head
    flops_fwd += 2 * batch_size * seq_length * hidden_size * vocab_size
    # bwd = 2x fwd; full recompute adds another fwd
    mult = 4 if recompute else 3
    return flops_fwd * mult


def speed_metrics(
    split: str,
    start_time: float,
    num_samples: Optional[int] = None,
    num_steps: Optional[int] = None,
    num_tokens: Optional[int] = None,
    model_flops: Optional[float] = None,
) -> Dict[str, float]:
    """Reference: trainer_utils.py:351-379 — reports
    *_tokens_per_second_per_device and *_hardware_tflops_per_device."""
    runtime = time.time() - start_time
    result = {f"{split}_runtime": round(runtime, 4)}
    if runtime == 0:
        return result
    if num_samples is not None:
        result[f"{split}_samples_per_second"] = round(num_samples / runtime, 3)
    if num_steps is not None:
        result[f"{split}_steps_per_second"] = round(num_steps / runtime, 3)
    if num_tokens is not None:
        result[f"{split}_tokens_per_second_per_device"] = round(num_tokens / runtime, 2)
    if model_flops is not None:
        tflops = model_flops / runtime / 1e12
        result[f"{split}_hardware_tflops_per_device"] = round(tflops, 2)
        result[f"{split}_mfu_percent"] = round(
            100.0 * model_flops / runtime / MI355X_BF16_PEAK_FLOPS, 2
        )
    return result


# ---------------------------------------------------------------------------
# LR schedulers (reference: paddlenlp/transformers/optimization.py)
# ---------------------------------------------------------------------------
def get_scheduler(
    name: str,
    optimizer: torch.optim.Optimizer,
    num_warmup_steps: int,
    num_training_steps: int,
    min_lr_ratio: float = 0.0,
    last_epoch: int = -1,
):
    name = name.lower()

    def linear(step):
        if step < num_warmup_steps:
            return step / max(1, num_warmup_steps)
        return max(
            min_lr_ratio,
            (num_training_steps - step) / max(1, num_training_steps - num_warmup_steps),
        )

    def cosine(step):
        if step < num_warmup_steps:
            return step / max(1, num_warmup_steps)
        progress = (step - num_warmup_steps) / max(1, num_training_steps - num_warmup_steps)
        progress = min(progress, 1.0)
        return min_lr_ratio + (1 - min_lr_ratio) * 0.5 * (1 + math.cos(math.pi * progress))

    def constant(step):
        if step < num_warmup_steps:
            return step / max(1, num_warmup_steps)
        return 1.0

    fns = {
        "linear": linear,
        "cosine": cosine,
        "constant": constant,
        "constant_with_warmup": constant,
    }
    if name not in fns:
        raise ValueError(f"Unknown scheduler {name}; choices: {sorted(fns)}")
    return torch.optim.lr_scheduler.LambdaLR(optimizer, fns[name], last_epoch=last_epoch)


CHECKPOINT_DONE_MARKER = ".checkpoint_done"


def get_last_checkpoint(folder: str) -> Optional[str]:
    """Latest COMPLETE checkpoint: only directories carrying the
    `.checkpoint_done` integrity marker qualify (reference PDC marker,
    trainer.py:2520-2524), so a save interrupted mid-write is never
    resumed from.  Falls back to the newest unmarked checkpoint only when
    no marked one exists (pre-marker checkpoints)."""
    if not os.path.isdir(folder):
        return None
    pat = re.compile(rf"^{PREFIX_CHECKPOINT_DIR}-(\d+)$")
    ckpts = [
        d for d in os.listdir(folder)
        if pat.match(d) and os.path.isdir(os.path.join(folder, d))
    ]
    if not ckpts:
        return None
    done = [d for d in ckpts
            if os.path.isfile(os.path.join(folder, d, CHECKPOINT_DONE_MARKER))]
    pool = done or ckpts
    return os.path.join(folder, max(pool, key=lambda d: int(pat.match(d).group(1))))


class TrainOutput:
    def __init__(self, global_step: int, training_loss: float, metrics: Dict[str, float]):
        self.global_step = global_step
        self.training_loss = training_loss
        self.metrics = metrics


def should_skip_data(global_step: int, skip_data_intervals) -> bool:
    """True when global_step (1-based, the step ABOUT to run) falls inside any
    [start, end] interval (reference trainer.py:1007 should_skip_data) —
    used to jump over corrupted data ranges on resume."""
    if not skip_data_intervals:
        return False
    for interval in skip_data_intervals:
        if len(interval) != 2 or interval[0] > interval[1]:
            raise ValueError(f"invalid skip interval {interval}")
        if interval[0] <= global_step <= interval[1]:
            return True
    return False


class TrainerMemoryTracker:
    """CPU RSS + device peak memory deltas per stage (reference
    trainer_utils.py:726).  Usage: start() ... stop_and_update_metrics(m)."""

    def __init__(self, skip: bool = False):
        self.skip = skip
        self._stage = None
        try:
            import psutil

            self._proc = psutil.Process()
        except ImportError:  # psutil is present in this image; belt+braces
            self._proc = None
            self.skip = True

    def _cpu_mem(self) -> int:
        return self._proc.memory_info().rss if self._proc else 0

    def start(self, stage: str = "train"):
        if self.skip:
            return
        import torch

        self._stage = stage
        self._cpu_begin = self._cpu_mem()
        if torch.cuda.is_available():
            torch.cuda.reset_peak_memory_stats()
            self._gpu_begin = torch.cuda.memory_allocated()
        else:
            self._gpu_begin = None

    def stop_and_update_metrics(self, metrics: dict):
        if self.skip or self._stage is None:
            return
        import torch

        stage = self._stage
        metrics[f"{stage}_mem_cpu_rss"] = self._cpu_mem()
        metrics[f"{stage}_mem_cpu_delta"] = self._cpu_mem() - self._cpu_begin
        if self._gpu_begin is not None:
            peak = torch.cuda.max_memory_allocated()
            metrics[f"{stage}_mem_gpu_alloc_delta"] = (
                torch.cuda.memory_allocated() - self._gpu_begin)
            metrics[f"{stage}_mem_gpu_peaked_delta"] = peak - self._gpu_begin
        self._stage = None


def enable_determinism(seed: int = 42):
    """Bitwise-reproducible training mode for CI (reference determinism
    knobs FLAGS_cudnn_deterministic etc.): seeds everything, turns on
    torch deterministic algorithms, MIOpen determinism, and routes custom
    kernels with atomic reductions (rms_norm backward) to their
    deterministic fallbacks via PNLP_DETERMINISTIC=1."""
    import os

    set_seed(seed)
    os.environ["PNLP_DETERMINISTIC"] = "1"
    os.environ.setdefault("MIOPEN_FIND_MODE", "1")
    os.environ.setdefault("CUBLAS_WORKSPACE_CONFIG", ":4096:8")
    torch.use_deterministic_algorithms(True, warn_only=True)


class DynamicLossScaler:
    """Dynamic fp16 loss scaling (reference paddle AMP O2 scaler semantics):
    scale the loss up before backward, unscale grads before clipping/step,
    skip the step and halve the scale on inf/nan, grow after
    `growth_interval` consecutive good steps."""

    def __init__(self, init_scale: float = 2.0 ** 16, growth_factor: float = 2.0,
                 backoff_factor: float = 0.5, growth_interval: int = 2000,
                 min_scale: float = 1.0):
        self.scale = float(init_scale)
        self.growth_factor = growth_factor
        self.backoff_factor = backoff_factor
        self.growth_interval = growth_interval
        self.min_scale = min_scale
        self._good_steps = 0

    def scale_loss(self, loss):
        return loss * self.scale

    def unscale_and_check(self, params) -> bool:
        """Divide grads by the scale in place; returns True when all grads
        are finite (step may proceed)."""
        inv = 1.0 / self.scale
        finite = True
        for p in params:
            if p.grad is None:
                continue
            p.grad.mul_(inv)
            if finite and not torch.isfinite(p.grad).all():
                finite = False
        return finite

    def update(self, found_inf: bool):
        if found_inf:
            self.scale = max(self.min_scale, self.scale * self.backoff_factor)
            self._good_steps = 0
        else:
            self._good_steps += 1
            if self._good_steps >= self.growth_interval:
                self.scale *= self.growth_factor
                self._good_steps = 0

    def state_dict(self):
        return {"scale": self.scale, "good_steps": self._good_steps}

    def load_state_dict(self, sd):
        self.scale = sd["scale"]
        self._good_steps = sd["good_steps"]
