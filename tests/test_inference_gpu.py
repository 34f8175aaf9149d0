"""GPU numerics for the paged-KV inference kernels + engine parity on device."""
import math

import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def C():
    from paddlenlp_amd.ops.functional import _load_extension

    return _load_extension()


def test_paged_decode_attn(C):
    from paddlenlp_amd.experimental.fused_transformer import paged_decode_attn_ref

    torch.manual_seed(0)
    B, Hq, Hk, D = 4, 8, 2, 128
    bs, nblocks, max_blocks = 16, 64, 8
    q = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
    k_cache = torch.randn(nblocks, bs, Hk, D, device="cuda", dtype=torch.bfloat16)
    v_cache = torch.randn(nblocks, bs, Hk, D, device="cuda", dtype=torch.bfloat16)
    # distinct random block tables per sequence
    perm = torch.randperm(nblocks)[: B * max_blocks].reshape(B, max_blocks)
    block_table = perm.to(torch.int32).cuda()
    seq_lens = torch.tensor([37, 5, 128, 64], dtype=torch.int32, device="cuda")

    out = C.paged_decode_attn(q, k_cache, v_cache, block_table, seq_lens)
    ref = paged_decode_attn_ref(q.cpu(), k_cache.cpu(), v_cache.cpu(),
                                block_table.cpu(), seq_lens.cpu())
    assert torch.allclose(out.cpu().float(), ref.float(), atol=3e-2, rtol=3e-2), \
        (out.cpu().float() - ref.float()).abs().max()


def test_rope_cache_append(C):
    from paddlenlp_amd.experimental.fused_transformer import rope_cache_append_ref
    from paddlenlp_amd import ops

    torch.manual_seed(1)
    B, T, Hq, Hk, D = 2, 6, 4, 2, 128
    bs, nblocks, max_blocks = 4, 32, 8
    qkv = torch.randn(B, T, (Hq + 2 * Hk) * D, device="cuda", dtype=torch.bfloat16)
    k_cache = torch.zeros(nblocks, bs, Hk, D, device="cuda", dtype=torch.bfloat16)
    v_cache = torch.zeros_like(k_cache)
    block_table = torch.arange(B * max_blocks, dtype=torch.int32).reshape(B, max_blocks).cuda()
    lens_before = torch.tensor([3, 0], dtype=torch.int32, device="cuda")
    counts = torch.tensor([6, 4], dtype=torch.int32, device="cuda")
    cos, sin = ops.build_rope_cache(64, D, device="cuda")

    q_out = C.rope_cache_append(qkv, k_cache, v_cache, block_table, lens_before,
                                cos, sin, Hq, Hk, counts)

    kc_ref = torch.zeros(nblocks, bs, Hk, D, dtype=torch.bfloat16)
    vc_ref = torch.zeros_like(kc_ref)
    q_ref = rope_cache_append_ref(qkv.cpu(), kc_ref, vc_ref, block_table.cpu(),
                                  lens_before.cpu(), cos.cpu(), sin.cpu(), Hq, Hk,
                                  counts.cpu())
    assert torch.allclose(q_out.cpu().float(), q_ref.float(), atol=2e-2, rtol=2e-2)
    assert torch.allclose(k_cache.cpu().float(), kc_ref.float(), atol=2e-2, rtol=2e-2), \
        (k_cache.cpu().float() - kc_ref.float()).abs().max()
    assert torch.allclose(v_cache.cpu().float(), vc_ref.float(), atol=2e-2, rtol=2e-2)


def test_engine_decode_gpu_matches_dygraph(C):
    from paddlenlp_amd.experimental import BlockManager, FusedMultiTransformer
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(3)
    cfg = LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256,
    )
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device="cuda")
    model.eval()
    eng = FusedMultiTransformer.from_llama(model, block_size=16, max_seq_len=256).to("cuda")
    eng.allocate_caches(num_blocks=64, device="cuda")
    B, T = 2, 24
    ids = torch.randint(3, 512, (B, T), device="cuda")
    lens = torch.tensor([T, T], dtype=torch.int32, device="cuda")
    mgr = BlockManager(64, 16, 16, B)
    slots = [mgr.allocate_slot(T) for _ in range(B)]
    bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)

    logits = eng.prefill(ids, bt, lens)
    with torch.no_grad():
        ref = model(input_ids=ids)[:, -1]
    # bf16 end-to-end: compare top-1 agreement + value closeness
    assert torch.allclose(logits, ref.float(), atol=0.5, rtol=5e-2), \
        (logits - ref.float()).abs().max()
    # bf16 end-to-end: the engine's chosen token must be near-optimal under
    # the dygraph logits (exact argmax can flip on sub-ulp ties)
    chosen = ref.float().gather(-1, logits.argmax(-1, keepdim=True)).squeeze(-1)
    assert (ref.float().max(-1).values - chosen < 0.25).all()

    # a few decode steps
    all_ids = ids.clone()
    for _ in range(3):
        nxt = logits.argmax(-1, keepdim=True)
        all_ids = torch.cat([all_ids, nxt], dim=1)
        lens_before = torch.tensor([int(mgr.seq_lens[s]) for s in slots],
                                   dtype=torch.int32, device="cuda")
        for s in slots:
            assert mgr.extend(s, 1)
        bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)
        logits = eng.decode_step(nxt, bt, lens_before)
        with torch.no_grad():
            ref = model(input_ids=all_ids)[:, -1]
        chosen = ref.float().gather(-1, logits.argmax(-1, keepdim=True)).squeeze(-1)
        assert (ref.float().max(-1).values - chosen < 0.25).all(), \
            (ref.float().max(-1).values - chosen).max()


def test_fp8_scaled_mm_gpu(C):
    """fp8 e4m3fn weight-only GEMM via the gfx950 fp8 MFMA path."""
    from paddlenlp_amd.quantization import QuantizationLinear

    torch.manual_seed(0)
    lin = torch.nn.Linear(256, 512, bias=False).to("cuda", torch.bfloat16)
    q = QuantizationLinear.from_linear(lin, "fp8")
    x = torch.randn(64, 256, device="cuda", dtype=torch.bfloat16)
    ref = lin(x)
    out = q(x)
    rel = (out.float() - ref.float()).abs().max() / ref.float().abs().max()
    assert rel < 0.15, rel


def test_engine_fp8_decode_gpu(C):
    from paddlenlp_amd.experimental import BlockManager, FusedMultiTransformer
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(3)
    cfg = LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128,
    )
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device="cuda")
    model.eval()
    eng = FusedMultiTransformer.from_llama(model, block_size=16, max_seq_len=128).to("cuda")
    eng.allocate_caches(num_blocks=32, device="cuda")
    eng.quantize("fp8")
    B, T = 2, 16
    ids = torch.randint(3, 512, (B, T), device="cuda")
    lens = torch.tensor([T, T], dtype=torch.int32, device="cuda")
    mgr = BlockManager(32, 16, 8, B)
    slots = [mgr.allocate_slot(T) for _ in range(B)]
    bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)
    logits = eng.prefill(ids, bt, lens)
    with torch.no_grad():
        ref = model(input_ids=ids)[:, -1]
    rel = (logits - ref.float()).abs().max() / ref.float().abs().max()
    assert torch.isfinite(logits).all()
    assert rel < 0.3, rel


def test_wint8_gemv_kernel(C):
    from paddlenlp_amd.quantization import quantize_int8

    torch.manual_seed(0)
    for M in (1, 3, 8, 16):
        x = torch.randn(M, 512, device="cuda", dtype=torch.bfloat16)
        w = torch.randn(384, 512, device="cuda", dtype=torch.bfloat16)
        q, sc = quantize_int8(w)
        y = C.wint8_gemv(x, q, sc)
        ref = x.float() @ (q.float() * sc[:, None]).t()
        rel = (y.float() - ref).abs().max() / ref.abs().max()
        assert rel < 0.02, (M, rel)


def test_graph_decode_runner_matches_eager(C):
    """hipGraph-captured decode must match eager decode bit-for-bit on the
    same inputs, including across changing seq lens and a batch-size switch."""
    import torch

    from paddlenlp_amd.experimental.block_manager import BlockManager
    from paddlenlp_amd.experimental.fused_transformer import GraphDecodeRunner
    from paddlenlp_amd.experimental import FusedMultiTransformer
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    device = torch.device("cuda:0")
    lcfg = LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256)
    model = LlamaForCausalLM.from_config(lcfg, dtype=torch.bfloat16,
                                         device="cuda").eval()
    eng = FusedMultiTransformer.from_llama(
        model, block_size=16, max_seq_len=256).to(device)
    eng.allocate_caches(64, device)
    runner = GraphDecodeRunner(eng)

    for B in (2, 4):
        mgr = BlockManager(64, 16, 8, B)
        prompt = 13
        ids = torch.randint(3, 512, (B, prompt), device=device)
        lens = torch.full((B,), prompt, dtype=torch.int32, device=device)
        slots = [mgr.allocate_slot(prompt) for _ in range(B)]
        bt = torch.stack([mgr.block_table[s] for s in slots]).to(device, torch.int32)
        eng.prefill(ids, bt, lens)
        tok = torch.randint(3, 512, (B, 1), device=device)
        for step in range(3):
            lens_before = torch.tensor(
                [int(mgr.seq_lens[s]) for s in slots], dtype=torch.int32,
                device=device)
            for s in slots:
                assert mgr.extend(s, 1)
            bt = torch.stack([mgr.block_table[s] for s in slots]).to(
                device, torch.int32)
            eager = eng.decode_step(tok, bt, lens_before)
            graphed = runner(tok, bt, lens_before)
            torch.testing.assert_close(graphed, eager, rtol=0, atol=0)
            # the cache was appended TWICE (eager + graph) for the same
            # position - identical values, so attention output is unchanged;
            # advance state once
            tok = eager.argmax(-1, keepdim=True)
        # free for the next batch size
        for s in slots:
            mgr.release(s)


def test_int8_kv_cache_kernels(C):
    """int8 cache kernels (quantized append + dequantizing decode) match the
    CPU int8 reference and stay close to the bf16 kernels."""
    import torch

    from paddlenlp_amd.experimental.fused_transformer import (
        paged_decode_attn_ref, rope_cache_append_ref)
    from paddlenlp_amd.ops import reference

    torch.manual_seed(0)
    device = torch.device("cuda:0")
    B, T, Hq, Hk, D = 2, 6, 4, 2, 128
    nblocks, bs, max_blocks = 8, 8, 4
    qkv = torch.randn(B, T, (Hq + 2 * Hk) * D, device=device,
                      dtype=torch.bfloat16)
    bt = torch.arange(B * max_blocks, device=device,
                      dtype=torch.int32).reshape(B, max_blocks)
    lens0 = torch.zeros(B, dtype=torch.int32, device=device)
    cos, sin = reference.build_rope_cache(64, D, 10000.0)
    cos = cos.to(device)
    sin = sin.to(device)

    kc = torch.zeros(nblocks, bs, Hk, D, dtype=torch.int8, device=device)
    vc = torch.zeros_like(kc)
    ks = torch.zeros(nblocks, bs, Hk, dtype=torch.float32, device=device)
    vs = torch.zeros_like(ks)
    q_out = C.rope_cache_append(qkv, kc, vc, bt, lens0, cos, sin, Hq, Hk,
                                None, ks, vs)

    kc_ref = torch.zeros_like(kc, device="cpu")
    vc_ref = torch.zeros_like(vc, device="cpu")
    ks_ref = torch.zeros_like(ks, device="cpu")
    vs_ref = torch.zeros_like(vs, device="cpu")
    q_ref = rope_cache_append_ref(
        qkv.cpu(), kc_ref, vc_ref, bt.cpu(), lens0.cpu(), cos.cpu(), sin.cpu(),
        Hq, Hk, None, ks_ref, vs_ref)
    torch.testing.assert_close(q_out.cpu().float(), q_ref.float(),
                               rtol=2e-2, atol=2e-2)
    # quantized cache contents match (int8 rounding is deterministic)
    assert (kc.cpu() - kc_ref).abs().max() <= 1
    torch.testing.assert_close(ks.cpu(), ks_ref, rtol=1e-3, atol=1e-5)

    # decode against the int8 cache
    q = torch.randn(B, Hq, D, device=device, dtype=torch.bfloat16)
    seq_lens = torch.full((B,), T, dtype=torch.int32, device=device)
    out = C.paged_decode_attn(q, kc, vc, bt, seq_lens, ks, vs)
    out_ref = paged_decode_attn_ref(q.cpu(), kc_ref, vc_ref, bt.cpu(),
                                    seq_lens.cpu(), ks_ref, vs_ref)
    rel = (out.cpu().float() - out_ref.float()).abs().max() \
        / out_ref.float().abs().max()
    assert rel < 0.03, rel


# ---------------------------------------------------------------------------
# fused sampling + device block scheduler (sampling.hip)
# ---------------------------------------------------------------------------
def test_topp_sample_greedy_matches_argmax(C):
    torch.manual_seed(0)
    B, V = 16, 1000
    logits = torch.randn(B, V, device="cuda", dtype=torch.bfloat16) * 4
    temp = torch.ones(B, device="cuda")
    top_p = torch.zeros(B, device="cuda")     # greedy
    u = torch.rand(B, device="cuda")
    out = C.topp_sample(logits, temp, top_p, u)
    assert torch.equal(out, logits.float().argmax(-1))


def test_topp_sample_membership(C):
    """Sampled tokens always land inside the torch-computed top-p set
    (allowing the documented 1/64-of-a-bin boundary slack)."""
    torch.manual_seed(1)
    B, V = 32, 2000
    logits = torch.randn(B, V, device="cuda", dtype=torch.bfloat16) * 3
    temp = torch.full((B,), 0.9, device="cuda")
    tp = 0.7
    top_p = torch.full((B,), tp, device="cuda")
    # torch reference top-p mask (slightly widened for boundary slack)
    lf = logits.float() / 0.9
    sorted_l, sorted_i = torch.sort(lf, descending=True)
    probs = sorted_l.softmax(-1)
    cum = probs.cumsum(-1)
    keep_sorted = (cum - probs) <= tp + 0.02
    keep = torch.zeros_like(keep_sorted).scatter(1, sorted_i, keep_sorted)
    for trial in range(8):
        u = torch.rand(B, device="cuda")
        out = C.topp_sample(logits, temp, top_p, u)
        ok = keep.gather(1, out.unsqueeze(1)).squeeze(1)
        assert ok.all(), (trial, out[~ok])


def test_topp_sample_min_length_bans_eos(C):
    torch.manual_seed(2)
    B, V = 4, 100
    logits = torch.full((B, V), -5.0, device="cuda", dtype=torch.bfloat16)
    logits[:, 7] = 10.0          # eos would win
    logits[:, 3] = 5.0
    temp = torch.ones(B, device="cuda")
    top_p = torch.zeros(B, device="cuda")
    u = torch.rand(B, device="cuda")
    eos = torch.tensor([7], dtype=torch.int64, device="cuda")
    cur = torch.zeros(B, dtype=torch.int32, device="cuda")
    minl = torch.full((B,), 5, dtype=torch.int32, device="cuda")
    out = C.topp_sample(logits, temp, top_p, u, eos, cur, minl)
    assert (out == 3).all(), out


def test_repetition_penalty_kernel(C):
    torch.manual_seed(3)
    B, V, L = 4, 500, 10
    logits = torch.randn(B, V, device="cuda", dtype=torch.bfloat16)
    ref = logits.float().clone()
    pre = torch.randint(0, V, (B, L), device="cuda", dtype=torch.int64)
    lens = torch.full((B,), L, dtype=torch.int32, device="cuda")
    rp = torch.full((B,), 1.3, device="cuda")
    C.apply_repetition_penalty(logits, pre, lens, rp)
    for b in range(B):
        for t in set(pre[b].tolist()):
            l = ref[b, t]
            ref[b, t] = l * 1.3 if l < 0 else l / 1.3
    assert torch.allclose(logits.float(), ref, atol=2e-2, rtol=2e-2)


def test_block_step_kernel_free_alloc_preempt(C):
    B, bs, mb, nblocks = 4, 16, 8, 10
    dev = "cuda"
    bt = torch.full((B, mb), -1, dtype=torch.int32, device=dev)
    # seq0: stopped with 2 blocks; seq1: active at a block boundary (needs
    # one); seq2: active mid-block; seq3: inactive
    bt[0, :2] = torch.tensor([5, 6], dtype=torch.int32)
    bt[1, :2] = torch.tensor([0, 1], dtype=torch.int32)
    bt[2, 0] = 2
    seq_lens = torch.tensor([20, 32, 5, 0], dtype=torch.int32, device=dev)
    stop = torch.tensor([1, 0, 0, 0], dtype=torch.int8, device=dev)
    active = torch.tensor([1, 1, 1, 0], dtype=torch.int8, device=dev)
    free_list = torch.tensor([9, 8, 7], dtype=torch.int32, device=dev)
    free_list = torch.cat([free_list, torch.zeros(nblocks - 3, dtype=torch.int32, device=dev)])
    free_count = torch.tensor([3], dtype=torch.int32, device=dev)
    is_bs = torch.zeros(B, dtype=torch.int8, device=dev)
    C.block_step(bt, seq_lens, stop, active, free_list, free_count, is_bs, bs)
    torch.cuda.synchronize()
    # seq0 freed (2 blocks back), inactive now
    assert int(active[0]) == 0 and int(seq_lens[0]) == 0
    assert (bt[0] == -1).all()
    # seq1 got a new block at position 2
    assert int(bt[1, 2]) >= 0
    # free count: 3 + 2 freed - 1 allocated = 4
    assert int(free_count[0]) == 4
    assert int(is_bs.sum()) == 0


def test_device_decode_loop_matches_host_loop(C):
    """Greedy device-resident loop == host-managed loop, token for token."""
    from paddlenlp_amd.experimental import BlockManager, FusedMultiTransformer
    from paddlenlp_amd.experimental.device_scheduler import DeviceDecodeLoop
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(5)
    cfg = LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256,
    )
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device="cuda")
    model.eval()
    eng = FusedMultiTransformer.from_llama(model, block_size=16, max_seq_len=256).to("cuda")
    eng.allocate_caches(num_blocks=64, device="cuda")
    B, T, STEPS = 2, 24, 10
    ids = torch.randint(3, 512, (B, T), device="cuda")
    lens = torch.tensor([T] * B, dtype=torch.int32, device="cuda")

    # ---- host loop (bench_infer pattern) ----
    mgr = BlockManager(64, 16, 16, B)
    slots = [mgr.allocate_slot(T) for _ in range(B)]
    bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)
    logits = eng.prefill(ids, bt, lens)
    # cast to bf16 before argmax on both sides: the device loop samples on
    # bf16 logits, and fp32-vs-bf16 argmax can flip on sub-eps ties
    tok = logits.to(torch.bfloat16).float().argmax(-1, keepdim=True)
    host_tokens = [tok.squeeze(1).cpu().clone()]
    for _ in range(STEPS - 1):
        lens_before = torch.tensor([int(mgr.seq_lens[s]) for s in slots],
                                   dtype=torch.int32, device="cuda")
        for s in slots:
            assert mgr.extend(s, 1)
        bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)
        lg = eng.decode_step(tok, bt, lens_before)
        tok = lg.to(torch.bfloat16).float().argmax(-1, keepdim=True)
        host_tokens.append(tok.squeeze(1).cpu().clone())

    # ---- device loop (fresh caches to avoid stale KV) ----
    eng.allocate_caches(num_blocks=64, device="cuda")
    loop = DeviceDecodeLoop(eng, max_batch=B, num_blocks=64,
                            max_blocks_per_seq=16, device="cuda",
                            eos_ids=[], max_gen_len=64)
    blocks = [loop.allocate_for_prefill(T) for _ in range(B)]
    bt2 = torch.stack([torch.cat([b, torch.full((16 - b.numel(),), -1,
                                                dtype=torch.int32, device="cuda")])
                       for b in blocks])
    logits2 = eng.prefill(ids, bt2, lens)
    first = logits2.to(torch.bfloat16).float().argmax(-1)
    for i in range(B):
        loop.add_request(i, T, int(first[i]), top_p=0.0, blocks=blocks[i])
    loop.decode_steps(STEPS - 1)
    torch.cuda.synchronize()

    for i in range(B):
        dev_hist = [int(first[i])] + loop.pre_ids[i, :STEPS - 1].cpu().tolist()
        host_hist = [int(host_tokens[s][i]) for s in range(STEPS)]
        assert dev_hist == host_hist, (i, dev_hist, host_hist)


def test_mixed_prefill_decode_step(C):
    """mixed_step (prefill on a side stream overlapping decode) must produce
    exactly the same logits as running the two calls sequentially."""
    from paddlenlp_amd.experimental import BlockManager, FusedMultiTransformer
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(7)
    cfg = LlamaConfig(
        vocab_size=512, hidden_size=256, intermediate_size=512,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=256,
    )
    model = LlamaForCausalLM.from_config(cfg, dtype=torch.bfloat16, device="cuda")
    model.eval()
    eng = FusedMultiTransformer.from_llama(model, block_size=16, max_seq_len=256).to("cuda")
    eng.allocate_caches(num_blocks=96, device="cuda")
    mgr = BlockManager(96, 16, 16, 4)

    # running decode batch of 2
    T = 24
    ids = torch.randint(3, 512, (2, T), device="cuda")
    lens = torch.tensor([T, T], dtype=torch.int32, device="cuda")
    slots = [mgr.allocate_slot(T) for _ in range(2)]
    bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)
    logits = eng.prefill(ids, bt, lens)
    tok = logits.argmax(-1, keepdim=True)
    lens_before = torch.tensor([int(mgr.seq_lens[s]) for s in slots],
                               dtype=torch.int32, device="cuda")
    for s in slots:
        mgr.extend(s, 1)
    bt = torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)

    # incoming prompt batch of 2
    p_ids = torch.randint(3, 512, (2, 20), device="cuda")
    p_lens = torch.tensor([20, 15], dtype=torch.int32, device="cuda")
    p_slots = [mgr.allocate_slot(20) for _ in range(2)]
    p_bt = torch.stack([mgr.block_table[s] for s in p_slots]).to("cuda", torch.int32)

    d_mix, p_mix = eng.mixed_step(tok, bt, lens_before, p_ids, p_bt, p_lens)
    torch.cuda.synchronize()

    # sequential reference on fresh caches (same block ids, rebuilt state)
    eng.allocate_caches(num_blocks=96, device="cuda")
    eng.prefill(ids, torch.stack([mgr.block_table[s] for s in slots]).to("cuda", torch.int32)[:, :],
                lens)
    d_seq = eng.decode_step(tok, bt, lens_before)
    p_seq = eng.prefill(p_ids, p_bt, p_lens)
    assert torch.equal(d_mix, d_seq)
    assert torch.equal(p_mix, p_seq)


def test_int4_kv_cache_gpu(C):
    """int4 cache kernels (quantized rope-append + dequantizing MFMA decode)
    match the CPU reference mirror."""
    from paddlenlp_amd.experimental.fused_transformer import (
        paged_decode_attn_ref, rope_cache_append_ref)

    torch.manual_seed(4)
    B, T, Hq, Hk, D = 2, 12, 8, 2, 128
    bs, nblocks, mb = 16, 16, 4
    qkv = torch.randn(B, T, (Hq + 2 * Hk) * D, device="cuda",
                      dtype=torch.bfloat16)
    from paddlenlp_amd import ops
    cos, sin = ops.build_rope_cache(64, D, device="cuda")
    bt = torch.arange(B * mb, dtype=torch.int32, device="cuda").reshape(B, mb)
    lens0 = torch.zeros(B, dtype=torch.int32, device="cuda")
    counts = torch.tensor([T, T - 3], dtype=torch.int32, device="cuda")

    kc = torch.zeros(nblocks, bs, Hk, D // 2, dtype=torch.uint8, device="cuda")
    vc = torch.zeros_like(kc)
    ks = torch.zeros(nblocks, bs, Hk, dtype=torch.float32, device="cuda")
    vs = torch.zeros_like(ks)
    q = C.rope_cache_append(qkv, kc, vc, bt, lens0, cos, sin, Hq, Hk,
                            counts, ks, vs)

    kc_ref = torch.zeros_like(kc, device="cpu")
    vc_ref = torch.zeros_like(vc, device="cpu")
    ks_ref = torch.zeros_like(ks, device="cpu")
    vs_ref = torch.zeros_like(vs, device="cpu")
    q_ref = rope_cache_append_ref(qkv.cpu(), kc_ref, vc_ref, bt.cpu(),
                                  lens0.cpu(), cos.cpu(), sin.cpu(), Hq, Hk,
                                  token_counts=counts.cpu(),
                                  k_scale=ks_ref, v_scale=vs_ref)
    assert torch.allclose(q.cpu().float(), q_ref.reshape(B, T, Hq, D).float(),
                          atol=2e-2, rtol=2e-2)
    # packed nibbles agree up to rounding ties (each nibble within 1 step);
    # scales must match closely
    assert torch.allclose(ks.cpu(), ks_ref, atol=1e-4, rtol=1e-3)
    assert torch.allclose(vs.cpu(), vs_ref, atol=1e-4, rtol=1e-3)
    kd = (kc.cpu().int() - kc_ref.int()).abs()
    vd = (vc.cpu().int() - vc_ref.int()).abs()
    assert int(kd.max()) <= 17 and int(vd.max()) <= 17, (kd.max(), vd.max())

    # decode against the (GPU-written) int4 cache; the CPU mirror reads the
    # SAME cache — comparing decodes over independently rounded caches
    # amplifies nibble ties through the softmax
    seq_lens = counts.clone()
    qd = torch.randn(B, Hq, D, device="cuda", dtype=torch.bfloat16)
    out = C.paged_decode_attn(qd, kc, vc, bt, seq_lens, ks, vs)
    out_ref = paged_decode_attn_ref(qd.cpu(), kc.cpu(), vc.cpu(), bt.cpu(),
                                    seq_lens.cpu(), ks.cpu(), vs.cpu())
    assert torch.allclose(out.cpu().float(), out_ref.float(),
                          atol=5e-2, rtol=5e-2), \
        (out.cpu().float() - out_ref.float()).abs().max()
