"""Per-stage training timers.

Reference: paddlenlp/trainer/plugins/timer.py:23-120 — _Timer with device
synchronize around wall-clock, timers for read-data / forward-backward /
all-reduce / optimizer-step (started/stopped at trainer.py:959-1133),
printed per log interval (_print_timer :1353).
"""
from __future__ import annotations

import time
from typing import Dict, Optional

import torch

from ...utils.log import logger

_GLOBAL_TIMERS: Optional["Timers"] = None


class _Timer:
    def __init__(self, name: str):
        self.name = name
        self.elapsed_ = 0.0
        self.started = False
        self.start_time = 0.0
        self.count = 0

    def _sync(self):
        if torch.cuda.is_available():
            torch.cuda.synchronize()

    def start(self):
        if self.started:
            return
        self._sync()
        self.start_time = time.time()
        self.started = True

    def stop(self):
        if not self.started:
            return
        self._sync()
        self.elapsed_ += time.time() - self.start_time
        self.count += 1
        self.started = False

    def elapsed(self, reset: bool = True) -> float:
        e = self.elapsed_
        if reset:
            self.elapsed_ = 0.0
            self.count = 0
        return e

    def reset(self):
        self.elapsed_ = 0.0
        self.count = 0
        self.started = False


class Timers:
    def __init__(self):
        self.timers: Dict[str, _Timer] = {}

    def __call__(self, name: str) -> _Timer:
        if name not in self.timers:
            self.timers[name] = _Timer(name)
        return self.timers[name]

    def log(self, names=None, reset: bool = True) -> str:
        names = names or list(self.timers)
        parts = []
        for name in names:
            if name in self.timers:
                e = self.timers[name].elapsed(reset=reset)
                if e > 0:
                    parts.append(f"{name}: {e*1000:.1f}ms")
        msg = " | ".join(parts)
        if msg:
            logger.info(f"[timers] {msg}")
        return msg


def get_timers() -> Timers:
    global _GLOBAL_TIMERS
    if _GLOBAL_TIMERS is None:
        _GLOBAL_TIMERS = Timers()
    return _GLOBAL_TIMERS


def set_timers(timers: Optional[Timers]):
    global _GLOBAL_TIMERS
    _GLOBAL_TIMERS = timers
