"""QuantizationConfig (reference: paddlenlp/quantization/quantization_config.py).

Validates the weight quantization algorithm and carries the block sizes the
4-bit paths use.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

SUPPORTED_ALGOS = [
    "weight_only_int8", "weight_only_int4", "nf4", "fp4", "fp8", "llm.int8",
]


@dataclass
class QuantizationConfig:
    weight_quantize_algo: Optional[str] = None
    quant_type: Optional[str] = None
    weight_blocksize: int = 64
    weight_double_quant: bool = False
    weight_double_quant_block_size: int = 256
    llm_int8_threshold: float = 6.0
    ignore_modules: List[str] = field(default_factory=lambda: ["lm_head"])

    def __post_init__(self):
        if (self.weight_quantize_algo is not None
                and self.weight_quantize_algo not in SUPPORTED_ALGOS):
            raise ValueError(
                f"weight_quantize_algo:{self.weight_quantize_algo} not in "
                f"supported list {SUPPORTED_ALGOS}")

    @property
    def is_weight_quantize(self) -> bool:
        return self.weight_quantize_algo is not None
