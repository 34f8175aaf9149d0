"""Compression arguments (reference: paddlenlp/trainer/compression_args.py).

`strategy` is a '+'-joined list from {dynabert, ptq, qat, embeddings}.
"""
from __future__ import annotations

from dataclasses import dataclass, field
from typing import List, Optional

VALID_STRATEGIES = {"dynabert", "ptq", "qat", "embeddings"}


@dataclass
class CompressionArguments:
    output_dir: str = "compress"
    strategy: str = "dynabert+ptq"
    # dynabert
    width_mult_list: List[float] = field(default_factory=lambda: [0.75])
    num_train_epochs: int = 1
    learning_rate: float = 5e-5
    per_device_train_batch_size: int = 8
    # ptq
    algo_list: List[str] = field(default_factory=lambda: ["avg", "abs_max", "mse"])
    batch_num_list: List[int] = field(default_factory=lambda: [4])
    batch_size_list: List[int] = field(default_factory=lambda: [8])
    weight_quantize_type: str = "channel_wise_abs_max"
    activation_quantize_type: str = "moving_average_abs_max"
    # qat
    moving_rate: float = 0.9
    logging_steps: int = 10
    input_dtype: str = "int64"
    device: Optional[str] = None

    def __post_init__(self):
        self.strategies = [s.strip() for s in self.strategy.split("+") if s.strip()]
        unknown = set(self.strategies) - VALID_STRATEGIES
        if unknown:
            raise ValueError(
                f"unknown compression strategies {sorted(unknown)}; "
                f"valid: {sorted(VALID_STRATEGIES)}")
