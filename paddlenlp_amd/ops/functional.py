"""Dispatch layer for fused ops: HIP/CDNA4 kernels on GPU, torch reference on CPU.

Design rule (BASELINE.json north star): on a GPU box the hand-written gfx950
extension is MANDATORY — if a CUDA tensor reaches one of these ops and the
extension is missing, we raise instead of silently falling back to eager
PyTorch.  The CPU path uses ops.reference (plain fp32 torch) so plumbing
tests run without a GPU.
"""
from __future__ import annotations

from typing import Optional

import os

import torch

from . import reference

_C = None
_C_IMPORT_ERROR = None


def _load_extension():
    global _C, _C_IMPORT_ERROR
    if _C is not None:
        return _C
    try:
        from . import _C as ext  # built in-tree by setup.py build_ext --inplace
        _C = ext
    except ImportError as e:  # pragma: no cover - GPU-box only
        _C_IMPORT_ERROR = e
        raise RuntimeError(
            "paddlenlp_amd.ops._C (the gfx950 HIP extension) is not built. "
            "Run `python setup.py build_ext --inplace` (or __graft_entry__.build()). "
            f"Original error: {e}"
        ) from e
    return _C


def extension_available() -> bool:
    try:
        _load_extension()
        return True
    except RuntimeError:
        return False


# ---------------------------------------------------------------------------
# RMSNorm
# ---------------------------------------------------------------------------
class _RMSNormFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, eps):
        C = _load_extension()
        x = x.contiguous()
        y, invrms = C.rms_norm_fwd(x, weight, eps)
        ctx.save_for_backward(x, weight, invrms)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _load_extension()
        x, weight, invrms = ctx.saved_tensors
        dx, dw = C.rms_norm_bwd(dy.contiguous(), x, weight, invrms)
        return dx, dw, None


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    # PNLP_DETERMINISTIC=1: the fused backward reduces dw with fp32 atomics
    # (run-to-run nondeterministic ordering); route through the pure-torch
    # path for bitwise-reproducible CI runs (reference determinism knobs
    # FLAGS_cudnn_deterministic / FLAGS_embedding_deterministic).
    if x.is_cuda and os.environ.get("PNLP_DETERMINISTIC", "0") != "1":
        return _RMSNormFunction.apply(x, weight, eps)
    return reference.rms_norm(x, weight, eps)


# ---------------------------------------------------------------------------
# Rotary position embedding (fused, Llama rotate-half convention)
# ---------------------------------------------------------------------------
class _RopeFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, cos, sin):
        C = _load_extension()
        q_out, k_out = C.rope_fwd(q.contiguous(), k.contiguous(), cos, sin, False)
        ctx.save_for_backward(cos, sin)
        return q_out, k_out

    @staticmethod
    def backward(ctx, dq, dk):
        C = _load_extension()
        cos, sin = ctx.saved_tensors
        # rotation is orthogonal: bwd = rotate by -theta
        dq_in, dk_in = C.rope_fwd(dq.contiguous(), dk.contiguous(), cos, sin, True)
        return dq_in, dk_in, None, None


def fused_rope(q, k, cos, sin):
    """q: [B,S,Hq,D], k: [B,S,Hk,D], cos/sin: [S,D]."""
    if q.is_cuda:
        return _RopeFunction.apply(q, k, cos, sin)
    return reference.apply_rope(q, k, cos, sin)


build_rope_cache = reference.build_rope_cache


# ---------------------------------------------------------------------------
# SwiGLU
# ---------------------------------------------------------------------------
class _SwiGLUFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        C = _load_extension()
        x = x.contiguous()
        y = C.swiglu_fwd(x)
        ctx.save_for_backward(x)
        return y

    @staticmethod
    def backward(ctx, dy):
        C = _load_extension()
        (x,) = ctx.saved_tensors
        return C.swiglu_bwd(dy.contiguous(), x)


def swiglu(x: torch.Tensor, y: Optional[torch.Tensor] = None) -> torch.Tensor:
    """silu(gate) * up.  x = cat([gate, up], -1) when y is None."""
    if y is not None:
        x = torch.cat([x, y], dim=-1)
    if x.is_cuda:
        return _SwiGLUFunction.apply(x)
    return reference.swiglu(x)


# ---------------------------------------------------------------------------
# FlashAttention (CDNA4 MFMA kernel; [B,S,H,D] layout, GQA native)
# ---------------------------------------------------------------------------
class _FlashAttnFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, causal):
        C = _load_extension()
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = C.flash_attn_fwd(q, k, v, causal)
        ctx.save_for_backward(q, k, v, o, lse)
        ctx.causal = causal
        return o

    @staticmethod
    def backward(ctx, do):
        C = _load_extension()
        q, k, v, o, lse = ctx.saved_tensors
        dq, dk, dv = C.flash_attn_bwd(do.contiguous(), q, k, v, o, lse, ctx.causal)
        return dq, dk, dv, None


class _FlashMaskFunction(torch.autograd.Function):
    """FlashMask sparse-causal attention (reference flashmask_attention,
    llama/fusion_ops.py:218-238): key j visible to queries j <= i < start[j]."""

    @staticmethod
    def forward(ctx, q, k, v, startend):
        C = _load_extension()
        q, k, v = q.contiguous(), k.contiguous(), v.contiguous()
        o, lse = C.flashmask_attn_fwd(q, k, v, startend)
        ctx.save_for_backward(q, k, v, o, lse, startend)
        return o

    @staticmethod
    def backward(ctx, do):
        C = _load_extension()
        q, k, v, o, lse, startend = ctx.saved_tensors
        dq, dk, dv = C.flashmask_attn_bwd(do.contiguous(), q, k, v, o, lse, startend)
        return dq, dk, dv, None


def flash_attention(
    q: torch.Tensor,
    k: torch.Tensor,
    v: torch.Tensor,
    causal: bool = True,
    attn_mask: Optional[torch.Tensor] = None,
    startend_row_indices: Optional[torch.Tensor] = None,
) -> torch.Tensor:
    """q: [B,S,Hq,D]; k,v: [B,S,Hk,D].  Returns [B,S,Hq,D]."""
    if q.is_cuda and attn_mask is None:
        if startend_row_indices is None:
            return _FlashAttnFunction.apply(q, k, v, causal)
        if causal:
            # [B, h(=1), Skv, 1] -> [B, Skv] (per-head masks pending)
            se = startend_row_indices
            if se.dim() == 4:
                assert se.shape[1] == 1, "per-kv-head FlashMask pending"
                se = se[:, 0, :, 0]
            return _FlashMaskFunction.apply(q, k, v, se.contiguous())
    return reference.flash_attention(
        q, k, v, causal=causal, attn_mask=attn_mask,
        startend_row_indices=startend_row_indices,
    )


# ---------------------------------------------------------------------------
# Fused cross-entropy (chunked; avoids materializing fp32 [N, V] twice)
# ---------------------------------------------------------------------------
class _CrossEntropyFunction(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, ignore_index):
        C = _load_extension()
        loss, lse = C.cross_entropy_fwd(logits.contiguous(), labels, ignore_index)
        ctx.save_for_backward(logits, labels, lse)
        ctx.ignore_index = ignore_index
        return loss

    @staticmethod
    def backward(ctx, dloss):
        C = _load_extension()
        logits, labels, lse = ctx.saved_tensors
        dlogits = C.cross_entropy_bwd(dloss.contiguous(), logits, labels, lse, ctx.ignore_index)
        return dlogits, None, None


def cross_entropy(logits, labels, ignore_index: int = -100, reduction: str = "none"):
    """Per-token loss [N] from logits [N, V]; reduction applied on top.

    The fused gfx950 kernel handles bf16 logits with V % 8 == 0 (the LLM
    vocab shapes); anything else computes eager on the same device."""
    if logits.is_cuda and logits.dtype == torch.bfloat16 and logits.shape[-1] % 8 == 0:
        loss = _CrossEntropyFunction.apply(logits, labels, ignore_index)
    else:
        loss = reference.cross_entropy(logits, labels, ignore_index, reduction="none")
    if reduction == "none":
        return loss
    mask = labels != ignore_index
    n = mask.sum().clamp(min=1)
    if reduction == "mean":
        return loss.sum() / n
    return loss.sum()


# ---------------------------------------------------------------------------
# Fused AdamW (multi-tensor)
# ---------------------------------------------------------------------------
def fused_adamw(
    params, grads, exp_avgs, exp_avg_sqs, masters,
    lr: float, beta1: float, beta2: float, eps: float,
    weight_decay: float, step: int,
):
    """Multi-tensor AdamW.  `masters` may be None (fp32 params) or a list of
    fp32 master weights matching bf16 `params`."""
    if params and params[0].is_cuda:
        C = _load_extension()
        C.fused_adamw(
            list(params), list(grads), list(exp_avgs), list(exp_avg_sqs),
            list(masters) if masters is not None else [],
            lr, beta1, beta2, eps, weight_decay, step,
        )
        return
    for i, p in enumerate(params):
        reference.adamw_step(
            p, grads[i], exp_avgs[i], exp_avg_sqs[i],
            masters[i] if masters is not None else None,
            lr, beta1, beta2, eps, weight_decay, step,
        )
