from .lora_layers import (  # noqa: F401
    ColumnParallelLoRALinear,
    LoRALinear,
    RowParallelLoRALinear,
)
from .lora_model import LoRAConfig, LoRAModel  # noqa: F401
