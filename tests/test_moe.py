"""Mixtral MoE tests: routing math, aux loss, EP all-to-all parity (gloo)."""
import torch
import torch.distributed as dist

from tests.test_distributed import _run_workers


def tiny_mixtral_cfg(**kw):
    from paddlenlp_amd.transformers.mixtral import MixtralConfig

    base = dict(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, num_local_experts=4, num_experts_per_tok=2,
        dtype="float32",
    )
    base.update(kw)
    return MixtralConfig(**base)


def test_mixtral_forward_backward():
    from paddlenlp_amd.transformers.mixtral import MixtralForCausalLM

    torch.manual_seed(0)
    model = MixtralForCausalLM.from_config(tiny_mixtral_cfg())
    ids = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    loss, logits = model(input_ids=ids, labels=labels)
    loss.backward()
    assert torch.isfinite(loss)
    assert logits.shape == (2, 16, 128)
    # router got gradients
    assert model.mixtral.layers[0].block_sparse_moe.gate.weight.grad is not None


def test_moe_block_equals_dense_computation():
    """Sparse dispatch == brute-force per-token expert evaluation."""
    from paddlenlp_amd.transformers.mixtral import MixtralSparseMoeBlock

    torch.manual_seed(1)
    cfg = tiny_mixtral_cfg()
    block = MixtralSparseMoeBlock(cfg)
    x = torch.randn(2, 8, 64)
    out, router_logits = block(x)

    flat = x.reshape(-1, 64)
    probs = block.gate(flat).float().softmax(-1)
    topw, tope = probs.topk(2, dim=-1)
    topw = topw / topw.sum(-1, keepdim=True)
    ref = torch.zeros_like(flat)
    for t in range(flat.shape[0]):
        for j in range(2):
            ref[t] += topw[t, j].to(flat.dtype) * block.experts[tope[t, j]](flat[t:t + 1])[0]
    assert torch.allclose(out.reshape(-1, 64), ref, atol=1e-5), \
        (out.reshape(-1, 64) - ref).abs().max()


def test_load_balancing_loss():
    from paddlenlp_amd.transformers.mixtral import load_balancing_loss_func

    # uniform router: every token picks the same top-2 (tie-break), probs
    # 1/E each -> loss = (2 experts x 1 x 1/E) x E = top_k = 2
    logits = torch.zeros(100, 4)
    loss = load_balancing_loss_func([logits], 4, 2)
    assert abs(loss.item() - 2.0) < 1e-5
    # a maximally imbalanced router (all mass on one expert) scores higher
    skew = torch.full((100, 4), -10.0)
    skew[:, 0] = 10.0
    assert load_balancing_loss_func([skew], 4, 1).item() > 3.5


def test_generate_mixtral():
    from paddlenlp_amd.transformers.mixtral import MixtralForCausalLM

    torch.manual_seed(0)
    model = MixtralForCausalLM.from_config(tiny_mixtral_cfg(eos_token_id=2))
    model.eval()
    ids = torch.randint(3, 128, (1, 8))
    out, _ = model.generate(ids, max_new_tokens=6)
    assert out.shape == (1, 6)


def _w_ep_parity(rank, world):
    """EP=2 Mixtral block == single-process block with all experts."""
    from paddlenlp_amd.parallel.topology import init_parallel_env
    from paddlenlp_amd.transformers.mixtral import MixtralSparseMoeBlock

    topo = init_parallel_env(dp_degree=world, backend="gloo")
    torch.manual_seed(5)
    full_cfg = tiny_mixtral_cfg()
    full_block = MixtralSparseMoeBlock(full_cfg)  # 4 experts, ep=1

    ep_cfg = tiny_mixtral_cfg(expert_parallel_degree=world)
    ep_block = MixtralSparseMoeBlock(ep_cfg)      # 2 local experts
    # copy weights: rank r owns experts [2r, 2r+2)
    with torch.no_grad():
        ep_block.gate.weight.copy_(full_block.gate.weight)
        for le in range(ep_block.experts_per_rank):
            ge = rank * ep_block.experts_per_rank + le
            for name in ("w1", "w2", "w3"):
                getattr(ep_block.experts, name)[le].copy_(
                    getattr(full_block.experts, name)[ge])

    x = torch.randn(2, 8, 64, generator=torch.Generator().manual_seed(7), requires_grad=True)
    out, _ = ep_block(x)
    ref, _ = full_block(x.detach())
    assert torch.allclose(out, ref, atol=1e-5), (out - ref).abs().max()
    # gradients flow through the two all-to-alls
    out.sum().backward()
    assert x.grad is not None and torch.isfinite(x.grad).all()
    # expert params are flagged no_sync (dp grad-allreduce exclusion)
    assert all(getattr(p, "no_sync", False) for p in ep_block.experts.parameters())


def test_expert_parallel_parity():
    _run_workers(_w_ep_parity)


def _tiny_qwen2_moe(**kw):
    from paddlenlp_amd.transformers import Qwen2MoeConfig

    return Qwen2MoeConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=3, num_attention_heads=4, num_key_value_heads=2,
        num_experts=4, num_experts_per_tok=2, moe_intermediate_size=32,
        shared_expert_intermediate_size=64, max_position_embeddings=128, **kw)


def test_qwen2_moe_forward_backward_and_layout():
    from paddlenlp_amd.transformers import Qwen2MoeForCausalLM

    torch.manual_seed(0)
    m = Qwen2MoeForCausalLM(_tiny_qwen2_moe(mlp_only_layers=[1]))
    # layer 1 forced dense, others sparse
    assert not m.qwen2_moe.layers[1].is_sparse
    assert m.qwen2_moe.layers[0].is_sparse and m.qwen2_moe.layers[2].is_sparse
    ids = torch.randint(0, 128, (2, 16))
    loss, logits = m(input_ids=ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss) and logits.shape == (2, 16, 128)
    # shared expert gate gets gradients
    blk = m.qwen2_moe.layers[0].mlp
    assert blk.shared_expert_gate.weight.grad is not None
    # qwen2 attention carries bias terms
    assert m.qwen2_moe.layers[0].self_attn.qkv_proj.bias is not None


def test_qwen2_moe_shared_expert_contribution():
    """Zeroing the shared-expert gate must change the output (the shared
    path is live), and norm_topk_prob toggles routed weighting."""
    from paddlenlp_amd.transformers.qwen2_moe.modeling import Qwen2MoeSparseMoeBlock

    torch.manual_seed(1)
    cfg = _tiny_qwen2_moe()
    blk = Qwen2MoeSparseMoeBlock(cfg).eval()
    x = torch.randn(1, 8, cfg.hidden_size)
    with torch.no_grad():
        out1, _ = blk(x)
        blk.shared_expert_gate.weight.zero_()
        out2, _ = blk(x)
    assert not torch.allclose(out1, out2)


def test_qwen2_moe_cached_decode_parity():
    from paddlenlp_amd.transformers import Qwen2MoeForCausalLM

    torch.manual_seed(2)
    m = Qwen2MoeForCausalLM(_tiny_qwen2_moe()).eval()
    ids = torch.randint(0, 128, (1, 10))
    with torch.no_grad():
        full = m(input_ids=ids)
        logits, past = m(input_ids=ids[:, :-1], use_cache=True)
        step, _ = m(input_ids=ids[:, -1:], use_cache=True, past_key_values=past)
    torch.testing.assert_close(step[:, 0], full[:, -1], rtol=1e-3, atol=1e-3)


def test_grouped_experts_state_dict_roundtrip():
    """GroupedExperts keeps per-expert ModuleList-style keys so existing
    checkpoints and conversion mappings still apply."""
    from paddlenlp_amd.parallel.expert_parallel import GroupedExperts

    torch.manual_seed(11)
    ge = GroupedExperts(4, 16, 32)
    sd = ge.state_dict()
    assert "0.w1.weight" in sd and "3.w2.weight" in sd
    assert sd["0.w1.weight"].shape == (32, 16)   # nn.Linear [out, in] layout
    ge2 = GroupedExperts(4, 16, 32)
    ge2.load_state_dict(sd)
    for n in ("w1", "w2", "w3"):
        assert torch.equal(getattr(ge, n), getattr(ge2, n))
    # grouped forward == per-expert loop
    x = torch.randn(10, 16)
    counts = torch.tensor([3, 0, 5, 2])
    out = ge.forward_grouped(x, counts)
    start = 0
    for e, n in enumerate(counts.tolist()):
        if n:
            ref = ge[e](x[start:start + n])
            assert torch.allclose(out[start:start + n], ref, atol=1e-5)
        start += n
