"""Plain-PyTorch fp32 reference implementations of every fused op.

These serve two purposes:
  1. the CPU execution path (tests, plumbing configs — no GPU in CI), and
  2. the numerics oracle that GPU tests compare the HIP kernels against.

They are intentionally simple and written for clarity, not speed.  On a GPU
box the HIP extension (paddlenlp_amd.ops._C) is mandatory — see functional.py.

Reference op semantics follow PaddleNLP's fused ops:
fused_rms_norm, fused_rotary_position_embedding, swiglu, flash_attention,
fused cross-entropy (ParallelCrossEntropy's local form), fused AdamW.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F


def rms_norm(x: torch.Tensor, weight: torch.Tensor, eps: float = 1e-6) -> torch.Tensor:
    """y = x / rms(x) * weight, computed in fp32."""
    dtype = x.dtype
    x32 = x.float()
    variance = x32.pow(2).mean(-1, keepdim=True)
    y = x32 * torch.rsqrt(variance + eps)
    return (y * weight.float()).to(dtype)


def build_rope_cache(
    seq_len: int,
    head_dim: int,
    base: float = 10000.0,
    dtype: torch.dtype = torch.float32,
    device=None,
    position_ids: Optional[torch.Tensor] = None,
    scaling_factor: float = 1.0,
) -> Tuple[torch.Tensor, torch.Tensor]:
    """cos/sin tables of shape [seq_len, head_dim] (half-duplicated).

    Matches the Llama convention: inv_freq over even dims, table is
    cat(freqs, freqs) so that rotate_half applies to (x1, x2) halves.
    """
    inv_freq = 1.0 / (base ** (torch.arange(0, head_dim, 2, dtype=torch.float32, device=device) / head_dim))
    if position_ids is None:
        t = torch.arange(seq_len, dtype=torch.float32, device=device)
    else:
        t = position_ids.float()
    t = t / scaling_factor
    freqs = torch.outer(t, inv_freq)
    emb = torch.cat((freqs, freqs), dim=-1)
    return emb.cos().to(dtype), emb.sin().to(dtype)


def _rotate_half(x: torch.Tensor) -> torch.Tensor:
    x1 = x[..., : x.shape[-1] // 2]
    x2 = x[..., x.shape[-1] // 2:]
    return torch.cat((-x2, x1), dim=-1)


def apply_rope(
    q: torch.Tensor,  # [B, S, Hq, D]
    k: torch.Tensor,  # [B, S, Hk, D]
    cos: torch.Tensor,  # [S, D]
    sin: torch.Tensor,  # [S, D]
) -> Tuple[torch.Tensor, torch.Tensor]:
    cos = cos.to(q.dtype)[None, :, None, :]
    sin = sin.to(q.dtype)[None, :, None, :]
    q_out = q * cos + _rotate_half(q) * sin
    k_out = k * cos + _rotate_half(k) * sin
    return q_out, k_out


def swiglu(x: torch.Tensor, y: Optional[torch.Tensor] = None) -> torch.Tensor:
    """silu(gate) * up.  Single-arg form takes cat([gate, up], -1)."""
    if y is None:
        x, y = x.chunk(2, dim=-1)
    return F.silu(x) * y


def flash_attention(
    q: torch.Tensor,  # [B, S, Hq, D]
    k: torch.Tensor,  # [B, S, Hk, D]
    v: torch.Tensor,  # [B, S, Hk, D]
    causal: bool = True,
    attn_mask: Optional[torch.Tensor] = None,
    startend_row_indices: Optional[torch.Tensor] = None,
    return_lse: bool = False,
):
    """Reference attention in fp32 (materializes scores — small inputs only).

    GQA: k/v heads are repeated to match q heads.
    `startend_row_indices` is the FlashMask sparse-mask form
    (reference: flashmask_attention, llama/fusion_ops.py:218-238):
    [B, Hk_or_1, S, 1] int32 — for causal masking, key j attends to query
    rows i with j <= i < startend[j]; rows >= startend[j] are masked.
    """
    B, S, Hq, D = q.shape
    Hk = k.shape[2]
    if Hk != Hq:
        rep = Hq // Hk
        k = k.repeat_interleave(rep, dim=2)
        v = v.repeat_interleave(rep, dim=2)
    q32 = q.permute(0, 2, 1, 3).float()  # [B, H, S, D]
    k32 = k.permute(0, 2, 1, 3).float()
    v32 = v.permute(0, 2, 1, 3).float()
    scores = q32 @ k32.transpose(-1, -2) / math.sqrt(D)
    Skv = k32.shape[-2]
    if startend_row_indices is not None:
        # FlashMask: key column j is visible to query rows j <= i < start[j]
        start = startend_row_indices.to(torch.long)  # [B, h, Skv, 1]
        rows = torch.arange(S, device=q.device).view(1, 1, S, 1)
        cols_start = start.transpose(-1, -2)  # [B, h, 1, Skv]
        causal_m = torch.arange(Skv, device=q.device).view(1, 1, 1, Skv) <= rows
        visible = causal_m & (rows < cols_start)
        scores = scores.masked_fill(~visible, float("-inf"))
    elif causal:
        mask = torch.ones(S, Skv, dtype=torch.bool, device=q.device).tril(Skv - S)
        scores = scores.masked_fill(~mask, float("-inf"))
    if attn_mask is not None:
        scores = scores + attn_mask.float()
    probs = scores.softmax(-1)
    # fully-masked rows produce NaN via softmax(-inf row); zero them
    probs = torch.nan_to_num(probs, nan=0.0)
    out = probs @ v32  # [B, H, S, D]
    out = out.permute(0, 2, 1, 3).to(q.dtype)
    if return_lse:
        lse = torch.logsumexp(scores, dim=-1)  # [B, H, S]; -inf for masked rows
        return out, lse
    return out


def cross_entropy(
    logits: torch.Tensor,  # [N, V]
    labels: torch.Tensor,  # [N]
    ignore_index: int = -100,
    reduction: str = "none",
) -> torch.Tensor:
    return F.cross_entropy(
        logits.float(), labels, ignore_index=ignore_index, reduction=reduction
    )


@torch.no_grad()
def adamw_step(
    param: torch.Tensor,
    grad: torch.Tensor,
    exp_avg: torch.Tensor,
    exp_avg_sq: torch.Tensor,
    master: Optional[torch.Tensor],
    lr: float,
    beta1: float,
    beta2: float,
    eps: float,
    weight_decay: float,
    step: int,
) -> None:
    """Single-tensor AdamW in fp32 with optional bf16 param + fp32 master."""
    p32 = master if master is not None else param
    g32 = grad.float()
    exp_avg.mul_(beta1).add_(g32, alpha=1 - beta1)
    exp_avg_sq.mul_(beta2).addcmul_(g32, g32, value=1 - beta2)
    bias1 = 1 - beta1**step
    bias2 = 1 - beta2**step
    denom = (exp_avg_sq / bias2).sqrt_().add_(eps)
    p32.mul_(1 - lr * weight_decay)
    p32.addcdiv_(exp_avg / bias1, denom, value=-lr)
    if master is not None:
        param.copy_(p32.to(param.dtype))
