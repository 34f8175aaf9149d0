from .quantization_linear import (  # noqa: F401
    QuantizationLinear,
    quantize_fp8,
    quantize_int8,
    weight_only_linear,
)
