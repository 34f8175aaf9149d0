"""Model-level quantization: swap nn.Linear for QuantizationLinear.

Reference behavior: paddlenlp/quantization/quantization_utils.py:38
(replace_with_quantization_linear walking named sublayers, skipping
ignore_modules such as lm_head).
"""
from __future__ import annotations

import torch.nn as nn

from ..utils.log import logger
from .quantization_config import QuantizationConfig
from .quantization_linear import QuantizationLinear


def replace_with_quantization_linear(model: nn.Module,
                                     quantization_config: QuantizationConfig):
    """Replace every nn.Linear (except ignore_modules matches) with a
    QuantizationLinear quantized from its weights in place.  Returns the
    list of replaced module names."""
    cfg = quantization_config
    algo = cfg.weight_quantize_algo or cfg.quant_type
    assert algo is not None, "quantization_config names no algorithm"
    replaced = []
    for name, module in list(model.named_modules()):
        if not isinstance(module, nn.Linear):
            continue
        if any(skip in name for skip in cfg.ignore_modules):
            continue
        qlin = QuantizationLinear.from_linear(
            module, algo, block_size=cfg.weight_blocksize)
        parent_name, _, leaf = name.rpartition(".")
        parent = model.get_submodule(parent_name) if parent_name else model
        setattr(parent, leaf, qlin)
        replaced.append(name)
    logger.info(f"quantized {len(replaced)} linears with {algo}")
    return replaced
