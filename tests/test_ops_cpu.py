"""CPU reference-op sanity: shapes, autograd, parity with plain torch."""
import math

import pytest
import torch
import torch.nn.functional as F

from paddlenlp_amd import ops


def test_rms_norm_matches_manual():
    x = torch.randn(4, 16, 64)
    w = torch.randn(64)
    y = ops.rms_norm(x, w, 1e-6)
    ref = x / (x.pow(2).mean(-1, keepdim=True) + 1e-6).sqrt() * w
    assert torch.allclose(y, ref, atol=1e-5)


def test_rope_orthogonality():
    torch.manual_seed(0)
    B, S, H, D = 2, 8, 4, 16
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, 2, D)
    cos, sin = ops.build_rope_cache(S, D)
    q1, k1 = ops.fused_rope(q, k, cos, sin)
    # norms preserved per (pair) rotation
    assert torch.allclose(q1.norm(), q.norm(), atol=1e-4)
    # position 0 is identity
    assert torch.allclose(q1[:, 0], q[:, 0], atol=1e-6)


def test_swiglu():
    x = torch.randn(4, 32)
    y = ops.swiglu(x)
    g, u = x.chunk(2, -1)
    assert torch.allclose(y, F.silu(g) * u, atol=1e-6)


def test_flash_attention_matches_sdpa():
    torch.manual_seed(0)
    B, S, Hq, Hk, D = 2, 32, 4, 2, 16
    q = torch.randn(B, S, Hq, D)
    k = torch.randn(B, S, Hk, D)
    v = torch.randn(B, S, Hk, D)
    out = ops.flash_attention(q, k, v, causal=True)
    # torch sdpa reference
    qt = q.permute(0, 2, 1, 3)
    kt = k.repeat_interleave(2, dim=2).permute(0, 2, 1, 3)
    vt = v.repeat_interleave(2, dim=2).permute(0, 2, 1, 3)
    ref = F.scaled_dot_product_attention(qt, kt, vt, is_causal=True)
    ref = ref.permute(0, 2, 1, 3)
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()


def test_flash_attention_gqa_backward():
    torch.manual_seed(0)
    B, S, Hq, Hk, D = 1, 16, 4, 2, 8
    q = torch.randn(B, S, Hq, D, requires_grad=True)
    k = torch.randn(B, S, Hk, D, requires_grad=True)
    v = torch.randn(B, S, Hk, D, requires_grad=True)
    out = ops.flash_attention(q, k, v, causal=True)
    out.sum().backward()
    assert q.grad is not None and k.grad is not None and v.grad is not None
    assert torch.isfinite(q.grad).all()


def test_cross_entropy_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(16, 100)
    labels = torch.randint(0, 100, (16,))
    labels[3] = -100
    loss = ops.cross_entropy(logits, labels, reduction="mean")
    ref = F.cross_entropy(logits, labels, ignore_index=-100)
    assert torch.allclose(loss, ref, atol=1e-5)


def test_adamw_matches_torch():
    torch.manual_seed(0)
    p_ref = torch.randn(32, requires_grad=False)
    p = p_ref.clone()
    g = torch.randn(32)
    m = torch.zeros(32)
    v = torch.zeros(32)
    ops.fused_adamw([p], [g], [m], [v], None, 1e-3, 0.9, 0.999, 1e-8, 0.01, 1)

    ref_opt = torch.optim.AdamW([p_ref.requires_grad_()], lr=1e-3, betas=(0.9, 0.999),
                                eps=1e-8, weight_decay=0.01)
    p_ref.grad = g.clone()
    ref_opt.step()
    assert torch.allclose(p, p_ref.detach(), atol=1e-6), (p - p_ref).abs().max()


def test_flashmask_startend_row_indices():
    """FlashMask semantics: key j visible to queries j <= i < start[j]."""
    torch.manual_seed(0)
    B, S, H, D = 1, 8, 1, 8
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    # plain causal via startend == S everywhere
    idx = torch.full((B, 1, S, 1), S, dtype=torch.int32)
    out = ops.flash_attention(q, k, v, causal=True, startend_row_indices=idx)
    ref = ops.flash_attention(q, k, v, causal=True)
    assert torch.allclose(out, ref, atol=1e-5)


def test_fused_head_and_loss_fn():
    """Chunked head+CE == plain lm_head + cross_entropy (values and grads)."""
    from paddlenlp_amd.transformers.tensor_parallel_utils import fused_head_and_loss_fn

    torch.manual_seed(0)
    N, H, V = 20, 16, 64
    hidden = torch.randn(N, H, requires_grad=True)
    weight = torch.randn(V, H, requires_grad=True)
    labels = torch.randint(0, V, (N,))
    labels[3] = -100
    loss = fused_head_and_loss_fn(hidden, weight, labels, chunk_tokens=7)
    hr = hidden.detach().clone().requires_grad_()
    wr = weight.detach().clone().requires_grad_()
    ref = F.cross_entropy(hr @ wr.t(), labels, ignore_index=-100)
    assert torch.allclose(loss, ref, atol=1e-5), (loss, ref)
    loss.backward()
    ref.backward()
    assert torch.allclose(hidden.grad, hr.grad, atol=1e-5), (hidden.grad - hr.grad).abs().max()
    assert torch.allclose(weight.grad, wr.grad, atol=1e-4), (weight.grad - wr.grad).abs().max()
