"""QLoRA block-wise 4-bit weight quantization (nf4 / fp4, double-quant).

Reference behavior: paddlenlp/quantization/qlora.py:20-115
(qlora_weight_quantize / dequantize / linear over paddleslim's
quant_blockwise).  MI355X design: the codebook lookup + block scaling run as
plain torch ops (the dequantized weight feeds the hipBLASLt bf16 GEMM); the
base weights stay frozen 4-bit while only LoRA adapters train, which is what
makes 8B-class finetunes fit comfortably in a fraction of the 288 GB HBM3E.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch

# 4-bit NormalFloat codebook (QLoRA paper, appendix E): quantiles of N(0,1)
# normalized to [-1, 1].
NF4_CODE = torch.tensor([
    -1.0, -0.6961928009986877, -0.5250730514526367, -0.39491748809814453,
    -0.28444138169288635, -0.18477343022823334, -0.09105003625154495, 0.0,
    0.07958029955625534, 0.16093020141124725, 0.24611230194568634,
    0.33791524171829224, 0.44070982933044434, 0.5626170039176941,
    0.7229568362236023, 1.0,
])

# fp4 (e2m1) codebook normalized to [-1, 1]: {0, .5, 1, 1.5, 2, 3, 4, 6}/6
_FP4_POS = torch.tensor([0.0, 0.5, 1.0, 1.5, 2.0, 3.0, 4.0, 6.0]) / 6.0
FP4_CODE = torch.cat([-_FP4_POS.flip(0)[:-1], _FP4_POS])  # 15 distinct values

_CODEBOOKS = {"nf4": NF4_CODE, "fp4": FP4_CODE}


def _codebook(quant_algo: str, device) -> torch.Tensor:
    try:
        return _CODEBOOKS[quant_algo].to(device)
    except KeyError:
        raise ValueError(f"unsupported 4-bit algo {quant_algo!r} "
                         f"(expected one of {sorted(_CODEBOOKS)})") from None


def quant_blockwise(w: torch.Tensor, quant_algo: str = "nf4",
                    block_size: int = 64) -> Tuple[torch.Tensor, torch.Tensor]:
    """w (any shape, numel % block_size == 0) -> (packed uint8 [numel//2],
    fp32 per-block absmax [numel//block_size]).  Row-major flattening."""
    assert w.numel() % block_size == 0, (w.shape, block_size)
    code = _codebook(quant_algo, w.device)
    flat = w.detach().float().reshape(-1, block_size)
    absmax = flat.abs().amax(dim=1).clamp(min=1e-12)
    normed = flat / absmax[:, None]
    idx = (normed[..., None] - code).abs().argmin(-1).to(torch.uint8)
    idx = idx.reshape(-1)
    packed = (idx[0::2] << 4) | idx[1::2]
    return packed, absmax


def dequant_blockwise(packed: torch.Tensor, absmax: torch.Tensor,
                      quant_algo: str = "nf4", block_size: int = 64,
                      dtype: torch.dtype = torch.float32) -> torch.Tensor:
    """Inverse of quant_blockwise -> flat tensor [numel]."""
    code = _codebook(quant_algo, packed.device)
    idx = torch.stack([packed >> 4, packed & 0xF], dim=1).reshape(-1).long()
    vals = code[idx].reshape(-1, block_size) * absmax[:, None]
    return vals.reshape(-1).to(dtype)


def _double_quant_scales(absmax: torch.Tensor, block_size: int = 256):
    """int8-quantize the per-block absmax (QLoRA double quantization):
    absmax -> (int8 q, fp32 super-scale [n/block], fp32 mean)."""
    mean = absmax.mean()
    centered = absmax - mean
    pad = (-centered.numel()) % block_size
    if pad:
        centered = torch.cat([centered, centered.new_zeros(pad)])
    blocks = centered.reshape(-1, block_size)
    sscale = blocks.abs().amax(dim=1).clamp(min=1e-12) / 127.0
    q = torch.clamp(torch.round(blocks / sscale[:, None]), -127, 127).to(torch.int8)
    return q.reshape(-1)[:absmax.numel()], sscale, mean


def _double_dequant_scales(q: torch.Tensor, sscale: torch.Tensor,
                           mean: torch.Tensor, block_size: int = 256):
    pad = (-q.numel()) % block_size
    qf = q.float()
    if pad:
        qf = torch.cat([qf, qf.new_zeros(pad)])
    absmax = (qf.reshape(-1, block_size) * sscale[:, None]).reshape(-1)
    return absmax[:q.numel()] + mean


def qlora_weight_quantize(weight: torch.Tensor, quant_algo: str = "nf4",
                          double_quant: bool = False, block_size: int = 64,
                          double_quant_block_size: int = 256):
    """-> (packed uint8, state dict).  Reference qlora.py:20-53."""
    packed, absmax = quant_blockwise(weight, quant_algo, block_size)
    state = {"shape": tuple(weight.shape), "quant_algo": quant_algo,
             "block_size": block_size, "double_quant": double_quant}
    if double_quant:
        q, sscale, mean = _double_quant_scales(absmax, double_quant_block_size)
        state.update(qabsmax=q, absmax_scale=sscale, absmax_mean=mean,
                     double_quant_block_size=double_quant_block_size)
    else:
        state["absmax"] = absmax
    return packed, state


def qlora_weight_dequantize(packed: torch.Tensor, state: dict,
                            dtype: torch.dtype = torch.float32) -> torch.Tensor:
    if state["double_quant"]:
        absmax = _double_dequant_scales(
            state["qabsmax"], state["absmax_scale"], state["absmax_mean"],
            state["double_quant_block_size"])
    else:
        absmax = state["absmax"]
    flat = dequant_blockwise(packed, absmax, state["quant_algo"],
                             state["block_size"], dtype)
    return flat.reshape(state["shape"])


def qlora_weight_quantize_dequantize(weight: torch.Tensor, quant_algo="nf4",
                                     double_quant=False, block_size=64,
                                     double_quant_block_size=256):
    """Round-trip helper (reference qlora.py:71-95) — e.g. for QAT-style
    error analysis."""
    packed, state = qlora_weight_quantize(
        weight, quant_algo, double_quant, block_size, double_quant_block_size)
    return qlora_weight_dequantize(packed, state, weight.dtype)


def qlora_weight_linear(x: torch.Tensor, packed: torch.Tensor, state: dict,
                        bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """y = x @ dequant(packed)^T (+ bias).  Reference qlora.py:98-115."""
    w = qlora_weight_dequantize(packed, state, x.dtype)
    y = x @ w.t()
    if bias is not None:
        y = y + bias
    return y
