"""GPU numerics: every HIP kernel vs the plain-torch fp32 reference.

All tests are marked gpu and run on a real MI355X via gpurun.  Tolerances are
bf16-appropriate (inputs are bf16; reference math is fp32 on the same bf16
inputs)."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def C():
    from paddlenlp_amd.ops.functional import _load_extension

    return _load_extension()


def _bf16(x):
    return x.to(torch.bfloat16)


def test_mfma_layout_probe(C):
    """Asymmetric-input check of the presumed A/B/C fragment layouts
    (guide G9: symmetric inputs cannot detect transposes)."""
    torch.manual_seed(0)
    A = _bf16(torch.randn(16, 32, device="cuda"))
    B = _bf16(torch.randn(32, 16, device="cuda"))
    out = C.mfma_probe(A, B)
    ref = A.float() @ B.float()
    assert torch.allclose(out, ref, atol=2e-2, rtol=2e-2), (out - ref).abs().max()


def test_mfma32_layout_probe(C):
    """32x32x16 fragment layouts (the v2 flash kernel builds on these)."""
    torch.manual_seed(1)
    A = _bf16(torch.randn(32, 16, device="cuda"))
    B = _bf16(torch.randn(16, 32, device="cuda"))
    out = C.mfma32_probe(A, B)
    ref = A.float() @ B.float()
    assert torch.allclose(out, ref, atol=2e-2, rtol=2e-2), (out - ref).abs().max()


def test_permlane32_swap_semantics(C):
    """v_permlane32_swap_b32: new_a = {a.lo32, b.lo32}, new_b = {a.hi32, b.hi32}
    viewed per-lane: l<32: (a[l], a[l+32]); l>=32: (b[l-32], b[l])."""
    o0, o1 = C.permlane_probe()
    o0, o1 = o0.cpu(), o1.cpu()
    for l in range(64):
        if l < 32:
            assert o0[l] == l, (l, o0[l])
            assert o1[l] == l + 32, (l, o1[l])
        else:
            assert o0[l] == 1000 + l - 32, (l, o0[l])
            assert o1[l] == 1000 + l, (l, o1[l])


def test_flash_attention_fwd_bench_shape(C):
    """Forward numerics at the training bench shape (S=4096, GQA 32:8)."""
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    B, S, Hq, Hk, D = 1, 4096, 32, 8, 128
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda"))
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda"))
    v = _bf16(torch.randn(B, S, Hk, D, device="cuda"))
    out = ops.flash_attention(q, k, v, causal=True)
    ref = ops.reference.flash_attention(q.float(), k.float(), v.float(), causal=True)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()


def test_rms_norm_fwd_bwd(C):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    x = _bf16(torch.randn(8, 64, 512, device="cuda")).requires_grad_()
    w = _bf16(torch.randn(512, device="cuda")).requires_grad_()
    y = ops.rms_norm(x, w, 1e-6)
    # fp32 reference on the same bf16 inputs
    xr = x.detach().float().requires_grad_()
    wr = w.detach().float().requires_grad_()
    yr = ops.reference.rms_norm(xr, wr, 1e-6)
    assert torch.allclose(y.float(), yr, atol=2e-2, rtol=2e-2)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.float())
    assert torch.allclose(x.grad.float(), xr.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(w.grad.float(), wr.grad, atol=2e-1, rtol=5e-2)


def test_rope_fwd_bwd(C):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    B, S, Hq, Hk, D = 2, 64, 4, 2, 128
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda")).requires_grad_()
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    cos, sin = ops.build_rope_cache(S, D, device="cuda")
    q1, k1 = ops.fused_rope(q, k, cos, sin)
    qr = q.detach().float().requires_grad_()
    kr = k.detach().float().requires_grad_()
    q2, k2 = ops.reference.apply_rope(qr, kr, cos, sin)
    assert torch.allclose(q1.float(), q2, atol=2e-2, rtol=2e-2)
    assert torch.allclose(k1.float(), k2, atol=2e-2, rtol=2e-2)
    dq = torch.randn_like(q1)
    dk = torch.randn_like(k1)
    (q1 * dq).sum().backward()
    (q2 * dq.float()).sum().backward()
    assert torch.allclose(q.grad.float(), qr.grad, atol=2e-2, rtol=2e-2)


def test_swiglu_fwd_bwd(C):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    x = _bf16(torch.randn(128, 256, device="cuda")).requires_grad_()
    y = ops.swiglu(x)
    xr = x.detach().float().requires_grad_()
    yr = ops.reference.swiglu(xr)
    assert torch.allclose(y.float(), yr, atol=2e-2, rtol=2e-2)
    dy = torch.randn_like(y)
    y.backward(dy)
    yr.backward(dy.float())
    assert torch.allclose(x.grad.float(), xr.grad, atol=3e-2, rtol=3e-2)


@pytest.mark.parametrize("B,Sq,Skv,Hq,Hk,D,causal", [
    (2, 128, 128, 4, 4, 128, True),
    (2, 128, 128, 4, 1, 128, True),     # GQA
    (1, 256, 256, 8, 2, 128, True),     # GQA 4:1
    (1, 100, 100, 2, 2, 128, True),     # ragged seq (not multiple of 64)
    (1, 128, 128, 2, 2, 64, True),      # head dim 64
    (2, 128, 128, 4, 4, 128, False),    # non-causal
    (1, 64, 192, 2, 2, 128, True),      # Skv > Sq (cached decode pattern)
    (1, 128, 64, 2, 2, 128, False),     # Sq > Skv non-causal (zigzag CP block)
    (1, 64, 128, 2, 2, 128, False),     # Skv > Sq non-causal (zigzag CP block)
])
def test_flash_attention_fwd(C, B, Sq, Skv, Hq, Hk, D, causal):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    q = _bf16(torch.randn(B, Sq, Hq, D, device="cuda"))
    k = _bf16(torch.randn(B, Skv, Hk, D, device="cuda"))
    v = _bf16(torch.randn(B, Skv, Hk, D, device="cuda"))
    out = ops.flash_attention(q, k, v, causal=causal)
    ref = ops.reference.flash_attention(
        q.float(), k.float(), v.float(), causal=causal)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()


@pytest.mark.parametrize("B,S,Hq,Hk,D", [
    (1, 128, 4, 4, 128),
    (1, 128, 4, 2, 128),
    (1, 100, 2, 2, 128),
    (1, 300, 8, 2, 128),     # multi-block both grids, ragged
    (1, 2048, 32, 8, 128),   # bench-shape GQA ratio at depth
])
def test_flash_attention_bwd(C, B, S, Hq, Hk, D):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda")).requires_grad_()
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    v = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    out = ops.flash_attention(q, k, v, causal=True)
    dout = torch.randn_like(out)
    out.backward(dout)

    qr = q.detach().float().requires_grad_()
    kr = k.detach().float().requires_grad_()
    vr = v.detach().float().requires_grad_()
    ref = ops.reference.flash_attention(qr, kr, vr, causal=True)
    ref.backward(dout.float())

    assert torch.allclose(q.grad.float(), qr.grad, atol=5e-2, rtol=5e-2), \
        (q.grad.float() - qr.grad).abs().max()
    assert torch.allclose(k.grad.float(), kr.grad, atol=5e-2, rtol=5e-2), \
        (k.grad.float() - kr.grad).abs().max()
    assert torch.allclose(v.grad.float(), vr.grad, atol=5e-2, rtol=5e-2), \
        (v.grad.float() - vr.grad).abs().max()


def test_cross_entropy_fwd_bwd(C):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    N, V = 64, 1024
    logits = _bf16(torch.randn(N, V, device="cuda")).requires_grad_()
    labels = torch.randint(0, V, (N,), device="cuda")
    labels[5] = -100
    loss = ops.cross_entropy(logits, labels, reduction="mean")
    lr = logits.detach().float().requires_grad_()
    ref = ops.reference.cross_entropy(lr, labels, reduction="none")
    mask = labels != -100
    ref = ref.sum() / mask.sum()
    assert torch.allclose(loss.float(), ref, atol=1e-2, rtol=1e-2)
    loss.backward()
    ref.backward()
    assert torch.allclose(logits.grad.float(), lr.grad, atol=1e-2, rtol=5e-2)


def test_fused_adamw_bf16_master(C):
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    p32 = torch.randn(1000, device="cuda")
    p = _bf16(p32.clone())
    master = p.float()  # fp32 master
    g = _bf16(torch.randn(1000, device="cuda"))
    m = torch.zeros(1000, device="cuda")
    v = torch.zeros(1000, device="cuda")
    ops.fused_adamw([p], [g], [m], [v], [master], 1e-3, 0.9, 0.999, 1e-8, 0.01, 1)

    # fp32 reference from the same starting point
    mr = torch.zeros(1000, device="cuda")
    vr = torch.zeros(1000, device="cuda")
    pr = _bf16(p32.clone()).float()
    ops.reference.adamw_step(pr, g.float(), mr, vr, None, 1e-3, 0.9, 0.999, 1e-8, 0.01, 1)
    assert torch.allclose(master, pr, atol=1e-5), (master - pr).abs().max()
    assert torch.allclose(m, mr, atol=1e-5)
    assert torch.allclose(p.float(), pr, atol=1e-2)  # bf16 rounding of master


def test_tiny_llama_fwd_bwd_gpu(C):
    """End-to-end tiny model through every HIP op."""
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=2,
        max_position_embeddings=256,
    )
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device="cuda")
    ids = torch.randint(0, 512, (2, 128), device="cuda")
    labels = torch.randint(0, 512, (2, 128), device="cuda")
    loss, logits = model(input_ids=ids, labels=labels)
    loss.backward()
    assert torch.isfinite(loss)
    # loss should be ~ log(512) for random init
    assert 4.0 < loss.item() < 9.0
    for n, prm in model.named_parameters():
        assert prm.grad is not None and torch.isfinite(prm.grad).all(), n


def test_flashmask_attention_gpu(C):
    """FlashMask packed-attention kernel vs the fp32 reference (fwd + bwd)."""
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    B, S, Hq, Hk, D = 2, 128, 4, 2, 128
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda")).requires_grad_()
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    v = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    # two packed samples: boundaries at 50 and 128
    se = torch.empty(B, 1, S, 1, dtype=torch.int32, device="cuda")
    se[:, 0, :50, 0] = 50
    se[:, 0, 50:, 0] = 128
    out = ops.flash_attention(q, k, v, causal=True, startend_row_indices=se)
    qr = q.detach().float().requires_grad_()
    kr = k.detach().float().requires_grad_()
    vr = v.detach().float().requires_grad_()
    ref = ops.reference.flash_attention(qr, kr, vr, causal=True, startend_row_indices=se)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()
    do = torch.randn_like(out)
    out.backward(do)
    ref.backward(do.float())
    assert torch.allclose(q.grad.float(), qr.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(k.grad.float(), kr.grad, atol=5e-2, rtol=5e-2)
    assert torch.allclose(v.grad.float(), vr.grad, atol=5e-2, rtol=5e-2)


def test_fused_head_and_loss_fn_gpu(C):
    from paddlenlp_amd.transformers.tensor_parallel_utils import fused_head_and_loss_fn

    torch.manual_seed(0)
    N, H, V = 256, 128, 1024
    hidden = _bf16(torch.randn(N, H, device="cuda")).requires_grad_()
    weight = _bf16(torch.randn(V, H, device="cuda")).requires_grad_()
    labels = torch.randint(0, V, (N,), device="cuda")
    labels[5] = -100
    loss = fused_head_and_loss_fn(hidden, weight, labels, chunk_tokens=100)
    hr = hidden.detach().float().requires_grad_()
    wr = weight.detach().float().requires_grad_()
    import torch.nn.functional as F
    ref = F.cross_entropy(hr @ wr.t(), labels, ignore_index=-100)
    assert torch.allclose(loss.float(), ref, atol=2e-2, rtol=1e-2), (loss, ref)
    loss.backward()
    ref.backward()
    assert torch.allclose(hidden.grad.float(), hr.grad, atol=2e-2, rtol=5e-2)
    assert torch.allclose(weight.grad.float(), wr.grad, atol=2e-2, rtol=5e-2)


@pytest.mark.parametrize("M,N,K,ks", [(64, 512, 1024, 4), (17, 384, 512, 2)])
def test_skinny_gemm(C, M, N, K, ks):
    torch.manual_seed(0)
    x = _bf16(torch.randn(M, K, device="cuda"))
    w = _bf16(torch.randn(N, K, device="cuda"))
    y = C.skinny_gemm(x, w, ks)
    ref = x.float() @ w.float().t()
    assert torch.allclose(y.float(), ref, atol=0.5, rtol=2e-2), \
        (y.float() - ref).abs().max()


def test_fused_head_and_loss_gpu_parity(C):
    """The chunked head+CE autograd.Function on the GPU kernel path must
    match the unfused lm_head @ + cross_entropy composition (loss, dhidden
    and dweight)."""
    from paddlenlp_amd.transformers.tensor_parallel_utils import (
        fused_head_and_loss_fn,
    )

    torch.manual_seed(0)
    N, H, V = 300, 64, 2048
    hidden = _bf16(torch.randn(N, H, device="cuda")).requires_grad_()
    weight = _bf16(torch.randn(V, H, device="cuda") * 0.02).requires_grad_()
    labels = torch.randint(0, V, (N,), device="cuda")
    labels[7] = -100

    loss = fused_head_and_loss_fn(hidden, weight, labels, chunk_tokens=128)
    loss.backward()

    h2 = hidden.detach().clone().requires_grad_()
    w2 = weight.detach().clone().requires_grad_()
    logits = h2 @ w2.t()
    ref = torch.nn.functional.cross_entropy(logits.float(), labels,
                                            ignore_index=-100)
    ref.backward()

    assert torch.allclose(loss.float(), ref.float(), atol=2e-2, rtol=2e-2), \
        (loss.item(), ref.item())
    assert torch.allclose(hidden.grad.float(), h2.grad.float(),
                          atol=2e-2, rtol=5e-2)
    assert torch.allclose(weight.grad.float(), w2.grad.float(),
                          atol=2e-2, rtol=5e-2)


def test_flash_attention_long_context_8k(C):
    """fwd+bwd numerics at S=8192 (long-context: 2x the bench shape, also
    exercises the XCD remap's fallback when gridDim.y % 8 != 0 via Hq=4)."""
    from paddlenlp_amd import ops

    torch.manual_seed(0)
    B, S, Hq, Hk, D = 1, 8192, 4, 2, 128
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda")).requires_grad_()
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    v = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    out = ops.flash_attention(q, k, v, causal=True)
    qr = q.detach().float().requires_grad_()
    kr = k.detach().float().requires_grad_()
    vr = v.detach().float().requires_grad_()
    ref = ops.reference.flash_attention(qr, kr, vr, causal=True)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()
    dy = torch.randn_like(out)
    out.backward(dy)
    ref.backward(dy.float())
    for g, gr, name in ((q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"),
                        (v.grad, vr.grad, "dv")):
        err = (g.float() - gr).abs().max().item()
        assert err < 8e-2, (name, err)


def test_flashmask_irregular_packing_s1024(C):
    """FlashMask v2 with tile-skipping: randomized, tile-unaligned sample
    boundaries across 16 kv tiles must match the fp32 reference (fwd+bwd)."""
    from paddlenlp_amd import ops

    torch.manual_seed(3)
    B, S, Hq, Hk, D = 2, 1024, 8, 2, 128
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda")).requires_grad_()
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    v = _bf16(torch.randn(B, S, Hk, D, device="cuda")).requires_grad_()
    # irregular boundaries, different per batch row
    se = torch.empty(B, 1, S, 1, dtype=torch.int32, device="cuda")
    for bi, bounds in enumerate(([37, 191, 500, 501, 777, 1024],
                                 [64, 640, 1024])):
        lo = 0
        for hi in bounds:
            se[bi, 0, lo:hi, 0] = hi
            lo = hi
    out = ops.flash_attention(q, k, v, causal=True, startend_row_indices=se)
    qr = q.detach().float().requires_grad_()
    kr = k.detach().float().requires_grad_()
    vr = v.detach().float().requires_grad_()
    ref = ops.reference.flash_attention(qr, kr, vr, causal=True,
                                        startend_row_indices=se)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()
    dy = torch.randn_like(out)
    out.backward(dy)
    ref.backward(dy.float())
    for g, gr, n in ((q.grad, qr.grad, "dq"), (k.grad, kr.grad, "dk"),
                     (v.grad, vr.grad, "dv")):
        err = (g.float() - gr).abs().max().item()
        assert err < 8e-2, (n, err)


def test_flashmask_bench_shape_gqa(C):
    """Masked v2 at the training bench geometry (S=4096, GQA 32:8, 8
    packed samples) — fwd vs fp32 reference."""
    from paddlenlp_amd import ops

    torch.manual_seed(5)
    B, S, Hq, Hk, D = 1, 4096, 32, 8, 128
    q = _bf16(torch.randn(B, S, Hq, D, device="cuda"))
    k = _bf16(torch.randn(B, S, Hk, D, device="cuda"))
    v = _bf16(torch.randn(B, S, Hk, D, device="cuda"))
    se = torch.empty(B, 1, S, 1, dtype=torch.int32, device="cuda")
    for i in range(8):
        se[:, 0, i * 512:(i + 1) * 512, 0] = (i + 1) * 512
    out = ops.flash_attention(q, k, v, causal=True, startend_row_indices=se)
    ref = ops.reference.flash_attention(q.float(), k.float(), v.float(),
                                        causal=True, startend_row_indices=se)
    assert torch.allclose(out.float(), ref, atol=3e-2, rtol=3e-2), \
        (out.float() - ref).abs().max()
