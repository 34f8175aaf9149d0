from .qlora import (  # noqa: F401
    qlora_weight_dequantize,
    qlora_weight_linear,
    qlora_weight_quantize,
    qlora_weight_quantize_dequantize,
)
from .quantization_config import QuantizationConfig  # noqa: F401
from .quantization_linear import (  # noqa: F401
    QuantizationLinear,
    quantize_fp8,
    quantize_int4,
    quantize_int8,
    weight_only_linear,
)
from .quantization_utils import replace_with_quantization_linear  # noqa: F401
