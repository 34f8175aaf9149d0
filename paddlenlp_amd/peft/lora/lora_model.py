"""LoRAModel + LoRAConfig.

Reference behavior: paddlenlp/peft/lora/lora_model.py (TP-aware save/merge,
mark_only_lora_as_trainable) and lora_config.py (target_modules regex list,
rslora, lora_plus_scale).
"""
from __future__ import annotations

import json
import os
import re
from dataclasses import asdict, dataclass, field
from typing import List

import torch.nn as nn

from ...parallel.tensor_parallel import ColumnParallelLinear, RowParallelLinear
from ...parallel.sequence_parallel import (
    ColumnSequenceParallelLinear,
    RowSequenceParallelLinear,
)
from ...quantization.quantization_linear import QuantizationLinear
from ...utils.log import logger
from .lora_layers import ColumnParallelLoRALinear, LoRALinear, RowParallelLoRALinear

LORA_CONFIG_NAME = "lora_config.json"
LORA_WEIGHTS_NAME = "lora_model_state.safetensors"


@dataclass
class LoRAConfig:
    r: int = 8
    lora_alpha: float = 16.0
    lora_dropout: float = 0.0
    target_modules: List[str] = field(default_factory=lambda: [
        ".*q_proj.*", ".*k_proj.*", ".*v_proj.*", ".*qkv_proj.*",
        ".*o_proj.*", ".*gate_proj.*", ".*up_proj.*", ".*down_proj.*",
        ".*gate_up_fused_proj.*",
    ])
    rslora: bool = False
    lora_plus_scale: float = 1.0
    merge_weights: bool = False

    def save_pretrained(self, path):
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, LORA_CONFIG_NAME), "w") as f:
            json.dump(asdict(self), f, indent=2)

    @classmethod
    def from_pretrained(cls, path):
        with open(os.path.join(path, LORA_CONFIG_NAME)) as f:
            return cls(**json.load(f))


class LoRAModel(nn.Module):
    def __init__(self, model: nn.Module, lora_config: LoRAConfig):
        super().__init__()
        self.model = model
        self.lora_config = lora_config
        self._inject(model, lora_config)
        self.mark_only_lora_as_trainable()

    @property
    def config(self):
        return getattr(self.model, "config", None)

    def _inject(self, model, cfg):
        patterns = [re.compile(p) for p in cfg.target_modules]
        replaced = 0
        for name, module in list(model.named_modules()):
            if not any(p.fullmatch(name) for p in patterns):
                continue
            kwargs = dict(r=cfg.r, lora_alpha=cfg.lora_alpha,
                          lora_dropout=cfg.lora_dropout, rslora=cfg.rslora,
                          lora_plus_scale=cfg.lora_plus_scale)
            if isinstance(module, (ColumnParallelLinear, ColumnSequenceParallelLinear)):
                new = ColumnParallelLoRALinear(module, **kwargs)
            elif isinstance(module, (RowParallelLinear, RowSequenceParallelLinear)):
                new = RowParallelLoRALinear(module, **kwargs)
            elif isinstance(module, (nn.Linear, QuantizationLinear)):
                new = LoRALinear(module, **kwargs)
            else:
                continue
            parent_name, _, leaf = name.rpartition(".")
            parent = model.get_submodule(parent_name) if parent_name else model
            setattr(parent, leaf, new)
            replaced += 1
        logger.info(f"LoRA: wrapped {replaced} linear layers (r={cfg.r})")
        if replaced == 0:
            raise ValueError(f"No modules matched target_modules {cfg.target_modules}")

    def mark_only_lora_as_trainable(self):
        for name, p in self.model.named_parameters():
            p.requires_grad_("lora_" in name)

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def generate(self, *args, **kwargs):
        return self.model.generate(*args, **kwargs)

    # ------------------------------------------------------------------
    def get_trainable_state_dict(self):
        return {k: v for k, v in self.model.state_dict().items() if "lora_" in k}

    def save_pretrained(self, path: str, merge_tensor_parallel: bool = False):
        from safetensors.torch import save_file

        os.makedirs(path, exist_ok=True)
        self.lora_config.save_pretrained(path)
        sd = {k: v.contiguous().cpu() for k, v in self.get_trainable_state_dict().items()}
        save_file(sd, os.path.join(path, LORA_WEIGHTS_NAME), metadata={"format": "pt"})
        logger.info(f"LoRA adapter saved to {path}")

    @classmethod
    def from_pretrained(cls, model: nn.Module, path: str):
        from safetensors.torch import load_file

        cfg = LoRAConfig.from_pretrained(path)
        lora = cls(model, cfg)
        sd = load_file(os.path.join(path, LORA_WEIGHTS_NAME))
        missing, unexpected = lora.model.load_state_dict(sd, strict=False)
        assert not unexpected, unexpected
        return lora

    def merge(self):
        for module in self.model.modules():
            if hasattr(module, "merge") and callable(module.merge) and hasattr(module, "lora_A"):
                module.merge()
