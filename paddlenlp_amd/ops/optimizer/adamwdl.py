"""AdamW with layer-wise LR decay (reference: paddlenlp/ops/optimizer/adamwdl.py)."""
from __future__ import annotations

import re
from typing import Callable, Optional

from ...trainer.optimizer import FusedAdamW


def _default_layer_fn(name: str, n_layers: int) -> int:
    m = re.search(r"layers?\.(\d+)\.", name)
    if m:
        return int(m.group(1)) + 1
    if "embed" in name:
        return 0
    return n_layers + 1


class AdamWDL(FusedAdamW):
    """Layer-wise learning-rate decay: lr_layer = lr * decay^(L - layer)."""

    def __init__(self, model, lr: float = 1e-4, layerwise_decay: float = 0.8,
                 n_layers: Optional[int] = None, name_to_layer: Optional[Callable] = None,
                 **kwargs):
        if n_layers is None:
            n_layers = getattr(getattr(model, "config", None), "num_hidden_layers", 12)
        fn = name_to_layer or (lambda n: _default_layer_fn(n, n_layers))
        buckets = {}
        for name, p in model.named_parameters():
            if not p.requires_grad:
                continue
            p.param_name = name
            layer = fn(name)
            buckets.setdefault(layer, []).append(p)
        groups = []
        for layer, params in sorted(buckets.items()):
            scale = layerwise_decay ** (n_layers + 1 - layer)
            groups.append({"params": params, "lr": lr * scale,
                           "weight_decay": kwargs.get("weight_decay", 0.0)})
        kwargs.pop("weight_decay", None)
        super().__init__(groups, lr=lr, **kwargs)
