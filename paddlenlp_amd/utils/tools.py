"""Small environment helpers (reference: paddlenlp/utils/tools.py
get_env_device and friends, re-targeted to torch/ROCm)."""
from __future__ import annotations

import torch


def get_env_device() -> str:
    """'gpu' on a visible ROCm/HIP device, else 'cpu' (reference
    get_env_device returns the paddle device string)."""
    if torch.cuda.is_available():
        return "gpu"
    return "cpu"


def device_name() -> str:
    if torch.cuda.is_available():
        return torch.cuda.get_device_name(0)
    return "cpu"


def is_gfx950() -> bool:
    """True on MI355X-class (CDNA4) devices."""
    if not torch.cuda.is_available():
        return False
    arch = getattr(torch.cuda.get_device_properties(0), "gcnArchName", "")
    return "gfx950" in arch


def compare_version(version: str, target: str) -> int:
    """-1 / 0 / 1 comparison of dotted version strings ('1.0' == '1.0.0')."""
    a = [int(x) for x in version.split(".")[:3] if x.isdigit()]
    b = [int(x) for x in target.split(".")[:3] if x.isdigit()]
    n = max(len(a), len(b))
    a += [0] * (n - len(a))
    b += [0] * (n - len(b))
    return (a > b) - (a < b)
