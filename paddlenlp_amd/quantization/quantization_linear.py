"""Weight-only quantized linear layers (int8 / fp8).

Reference behavior: paddlenlp/quantization/quantization_linear.py:42
(QuantizationLinear + Column/RowParallel variants :165/:277 — weight-only
int4/8 runtime) and the fp8 cutlass path (fused_transformer_layers.py
FusedMultiTransformerFP8 :2491).

MI355X design: fp8 uses OCP e4m3fn (the gfx950-native format — NOT the
MI300X fnuz variant) through torch._scaled_mm, which lowers to hipBLASLt's
fp8 MFMA path on gfx950 (~2x the bf16 rate).  int8 weight-only dequantizes
per-channel into the activation dtype and uses the bf16 GEMM; a fused
dequant-GEMM epilogue kernel is the planned upgrade.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn

FP8_E4M3_MAX = 448.0


def quantize_int8(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-output-channel symmetric int8: w [out, in] -> (int8 w, fp32 scale[out])."""
    scale = w.abs().amax(dim=1).clamp(min=1e-8).float() / 127.0
    q = torch.clamp(torch.round(w.float() / scale[:, None]), -127, 127).to(torch.int8)
    return q, scale


def quantize_int4(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-output-channel symmetric int4: w [out, in] -> (packed uint8
    [out, in//2] with two nibbles per byte, fp32 scale[out])."""
    assert w.shape[1] % 2 == 0, w.shape
    scale = w.abs().amax(dim=1).clamp(min=1e-8).float() / 7.0
    q = torch.clamp(torch.round(w.float() / scale[:, None]), -7, 7).to(torch.int8)
    u = (q + 8).to(torch.uint8)  # offset-binary nibbles
    packed = (u[:, 0::2] << 4) | u[:, 1::2]
    return packed, scale


def dequantize_int4(packed: torch.Tensor, scale: torch.Tensor,
                    dtype: torch.dtype) -> torch.Tensor:
    hi = (packed >> 4).to(torch.int8) - 8
    lo = (packed & 0xF).to(torch.int8) - 8
    q = torch.stack([hi, lo], dim=2).reshape(packed.shape[0], -1)
    return (q.float() * scale[:, None]).to(dtype)


def quantize_fp8(w: torch.Tensor) -> Tuple[torch.Tensor, torch.Tensor]:
    """Per-output-channel fp8 e4m3fn: w [out, in] -> (fp8 w, fp32 scale
    [out]).  Channelwise scales feed rowwise-scaled _scaled_mm (hipBLASLt
    fp8 MFMA) and quantize better than a single tensor scale."""
    scale = (w.abs().amax(dim=1).clamp(min=1e-8).float() / FP8_E4M3_MAX)
    q = (w.float() / scale[:, None]).clamp(-FP8_E4M3_MAX, FP8_E4M3_MAX).to(torch.float8_e4m3fn)
    return q, scale


def weight_only_linear(x: torch.Tensor, qweight: torch.Tensor, scale: torch.Tensor,
                       bias: Optional[torch.Tensor] = None,
                       algo: str = "weight_only_int8") -> torch.Tensor:
    """y = x @ dequant(qweight)^T (+ bias)."""
    if algo == "fp8":
        orig_shape = x.shape
        x2 = x.reshape(-1, orig_shape[-1])
        if x2.is_cuda:
            # fused one-kernel rowwise activation quant (per-token scale) +
            # rowwise/channelwise-scaled fp8 GEMM on the fp8 MFMA pipe
            from ..ops.functional import _load_extension

            C = _load_extension()
            x8, x_scale = C.fp8_rowwise_quant(x2.contiguous())
            if scale.dim() == 0:      # legacy per-tensor weight scale
                w_scale = scale.reshape(1, 1).expand(1, qweight.shape[0]).contiguous()
            else:
                w_scale = scale.reshape(1, -1).contiguous()
            y = torch._scaled_mm(
                x8, qweight.t(), scale_a=x_scale.unsqueeze(1),
                scale_b=w_scale, bias=None, out_dtype=x.dtype,
            )
        else:
            ws = scale if scale.dim() else scale.reshape(1)
            y = x2 @ (qweight.float() * (ws[:, None] if ws.dim() else ws)).t().to(x.dtype)
        y = y.reshape(*orig_shape[:-1], -1)
    elif algo == "weight_only_int4":
        w = dequantize_int4(qweight, scale, x.dtype)
        y = x @ w.t()
    elif algo == "weight_only_int8_gemv":
        # experimental fused dequant GEMV (ops/csrc/wint8_gemv.hip): correct
        # but currently VALU-bound (per-FMA LDS reads in the M loop) and
        # slower than dequant+hipBLASLt — kept for the register-tiled rework
        from ..ops.functional import _load_extension

        C = _load_extension()
        y = C.wint8_gemv(x, qweight, scale.float())
    else:
        w = (qweight.float() * scale[:, None]).to(x.dtype)
        y = x @ w.t()
    if bias is not None:
        y = y + bias
    return y


class QuantizationLinear(nn.Module):
    """Inference-only quantized linear (reference QuantizationLinear :42)."""

    def __init__(self, in_features: int, out_features: int,
                 quant_algo: str = "weight_only_int8", bias: bool = False,
                 dtype: torch.dtype = torch.bfloat16, block_size: int = 64):
        super().__init__()
        self.in_features = in_features
        self.out_features = out_features
        self.quant_algo = quant_algo
        self.block_size = block_size
        self.compute_dtype = dtype
        if quant_algo == "fp8":
            self.register_buffer("quant_weight",
                                 torch.zeros(out_features, in_features, dtype=torch.float8_e4m3fn))
            # per-output-channel scales (rowwise-scaled fp8 GEMM)
            self.register_buffer("quant_scale", torch.ones(out_features, dtype=torch.float32))
        elif quant_algo == "weight_only_int4":
            self.register_buffer("quant_weight",
                                 torch.zeros(out_features, in_features // 2, dtype=torch.uint8))
            self.register_buffer("quant_scale", torch.ones(out_features, dtype=torch.float32))
        elif quant_algo in ("nf4", "fp4"):
            n = out_features * in_features
            assert n % block_size == 0, (out_features, in_features, block_size)
            self.register_buffer("quant_weight", torch.zeros(n // 2, dtype=torch.uint8))
            self.register_buffer("quant_scale", torch.ones(n // block_size, dtype=torch.float32))
        else:
            self.register_buffer("quant_weight",
                                 torch.zeros(out_features, in_features, dtype=torch.int8))
            self.register_buffer("quant_scale", torch.ones(out_features, dtype=torch.float32))
        self.bias = nn.Parameter(torch.zeros(out_features, dtype=dtype)) if bias else None

    @classmethod
    def from_linear(cls, linear: nn.Linear, quant_algo: str = "weight_only_int8",
                    block_size: int = 64):
        m = cls(linear.in_features, linear.out_features, quant_algo,
                bias=linear.bias is not None, dtype=linear.weight.dtype,
                block_size=block_size)
        if quant_algo == "fp8":
            q, s = quantize_fp8(linear.weight.data)
        elif quant_algo == "weight_only_int4":
            q, s = quantize_int4(linear.weight.data)
        elif quant_algo in ("nf4", "fp4"):
            from .qlora import quant_blockwise
            q, s = quant_blockwise(linear.weight.data, quant_algo, block_size)
        else:
            q, s = quantize_int8(linear.weight.data)
        m.quant_weight.copy_(q)
        m.quant_scale.copy_(s)
        if linear.bias is not None:
            m.bias.data.copy_(linear.bias.data)
        m = m.to(linear.weight.device)
        return m

    def forward(self, x):
        if self.quant_algo in ("nf4", "fp4"):
            from .qlora import dequant_blockwise
            w = dequant_blockwise(self.quant_weight, self.quant_scale,
                                  self.quant_algo, self.block_size,
                                  x.dtype).reshape(self.out_features, self.in_features)
            y = x @ w.t()
            if self.bias is not None:
                y = y + self.bias
            return y
        return weight_only_linear(x, self.quant_weight, self.quant_scale,
                                  self.bias, self.quant_algo)
