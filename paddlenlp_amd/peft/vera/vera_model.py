"""VeRA: Vector-based Random-matrix Adaptation (reference: paddlenlp/peft/vera).

Shared frozen random A/B across layers; only per-layer scaling vectors
(lambda_d, lambda_b) train — far fewer parameters than LoRA.
"""
from __future__ import annotations

import json
import math
import os
from dataclasses import asdict, dataclass, field
from typing import List

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...utils.log import logger


@dataclass
class VeRAConfig:
    r: int = 256
    vera_alpha: float = 8.0
    target_modules: List[str] = field(default_factory=lambda: [
        ".*q_proj.*", ".*v_proj.*", ".*qkv_proj.*"])
    seed: int = 0


class VeRALinear(nn.Module):
    def __init__(self, base: nn.Linear, A: torch.Tensor, B: torch.Tensor, alpha: float, r: int):
        super().__init__()
        self.base = base
        in_f, out_f = base.in_features, base.out_features
        # shared frozen projections, sliced to this layer's shape
        self.register_buffer("vera_A", A[:r, :in_f].clone())
        self.register_buffer("vera_B", B[:out_f, :r].clone())
        self.lambda_d = nn.Parameter(torch.full((r,), 0.1, dtype=base.weight.dtype))
        self.lambda_b = nn.Parameter(torch.zeros(out_f, dtype=base.weight.dtype))
        self.scaling = alpha / r
        base.weight.requires_grad_(False)
        if base.bias is not None:
            base.bias.requires_grad_(False)

    def forward(self, x):
        h = F.linear(x, self.vera_A.to(x.dtype)) * self.lambda_d
        h = F.linear(h, self.vera_B.to(x.dtype)) * self.lambda_b
        return self.base(x) + self.scaling * h


class VeRAModel(nn.Module):
    def __init__(self, model: nn.Module, config: VeRAConfig):
        super().__init__()
        import re

        self.model = model
        self.vera_config = config
        g = torch.Generator().manual_seed(config.seed)
        max_in = max((m.in_features for m in model.modules() if isinstance(m, nn.Linear)), default=0)
        max_out = max((m.out_features for m in model.modules() if isinstance(m, nn.Linear)), default=0)
        A = torch.randn(config.r, max_in, generator=g) / math.sqrt(max_in)
        B = torch.randn(max_out, config.r, generator=g) / math.sqrt(config.r)
        patterns = [re.compile(p) for p in config.target_modules]
        replaced = 0
        for name, module in list(model.named_modules()):
            if isinstance(module, nn.Linear) and any(p.fullmatch(name) for p in patterns):
                parent_name, _, leaf = name.rpartition(".")
                parent = model.get_submodule(parent_name) if parent_name else model
                setattr(parent, leaf, VeRALinear(module, A, B, config.vera_alpha, config.r))
                replaced += 1
        if replaced == 0:
            raise ValueError(f"No modules matched {config.target_modules}")
        logger.info(f"VeRA: wrapped {replaced} linears (r={config.r})")
        for n, p in model.named_parameters():
            p.requires_grad_("lambda_" in n)

    @property
    def config(self):
        return getattr(self.model, "config", None)

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def save_pretrained(self, path):
        from safetensors.torch import save_file

        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, "vera_config.json"), "w") as f:
            json.dump(asdict(self.vera_config), f)
        sd = {k: v.detach().cpu().contiguous()
              for k, v in self.model.state_dict().items() if "lambda_" in k}
        save_file(sd, os.path.join(path, "vera_model_state.safetensors"),
                  metadata={"format": "pt"})
