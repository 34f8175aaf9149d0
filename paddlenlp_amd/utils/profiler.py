"""torch.profiler hook driven by the PROFILER_OPTIONS env variable.

Reference behavior: paddlenlp/utils/profiler.py (paddle profiler enabled by
PROFILER_OPTIONS="batch_range=[10, 20]; profile_path=model.profile").
MI355X note: this captures the HOST-side torch op timeline; for per-kernel
GPU time use rocprofv3 (profiles/ carries those captures).
"""
from __future__ import annotations

import os
from typing import Optional

import torch

_profiler = None
_batch_range = None
_path = None
_step = 0


def _parse_options(opts: str):
    out = {}
    for part in opts.split(";"):
        if "=" not in part:
            continue
        k, v = part.split("=", 1)
        out[k.strip()] = v.strip()
    return out


def add_profiler_step(options: Optional[str] = None):
    """Call once per train step; starts/steps/stops the profiler according to
    batch_range=[a, b] (inclusive) and writes a chrome trace to profile_path."""
    global _profiler, _batch_range, _path, _step
    options = options if options is not None else os.environ.get("PROFILER_OPTIONS")
    if not options:
        return
    if _batch_range is None:
        parsed = _parse_options(options)
        rng = parsed.get("batch_range", "[10, 20]").strip("[] ")
        lo, hi = (int(x) for x in rng.split(","))
        _batch_range = (lo, hi)
        _path = parsed.get("profile_path", "model.profile.json")
    _step += 1
    lo, hi = _batch_range
    if _step == lo and _profiler is None:
        activities = [torch.profiler.ProfilerActivity.CPU]
        if torch.cuda.is_available():
            activities.append(torch.profiler.ProfilerActivity.CUDA)
        _profiler = torch.profiler.profile(activities=activities)
        _profiler.__enter__()
    elif _profiler is not None and _step > hi:
        _profiler.__exit__(None, None, None)
        _profiler.export_chrome_trace(_path)
        _profiler = None
