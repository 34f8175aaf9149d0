"""DeepSeek-V2: MLA latent attention, group-limited top-k routing, shared
experts, dense-layer replacement.

Reference behavior: paddlenlp/transformers/deepseek_v2/modeling.py.
"""
import torch

from paddlenlp_amd.transformers import DeepseekV2Config, DeepseekV2ForCausalLM
from paddlenlp_amd.transformers.deepseek_v2.modeling import (
    DeepseekV2Attention,
    DeepseekV2MoE,
    DeepseekV2MoEGate,
)

torch.manual_seed(0)


def tiny_cfg(**kw):
    base = dict(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        moe_intermediate_size=32, num_hidden_layers=3, num_attention_heads=4,
        q_lora_rank=24, kv_lora_rank=16, qk_nope_head_dim=16,
        qk_rope_head_dim=8, v_head_dim=16, n_routed_experts=8,
        n_shared_experts=1, num_experts_per_tok=2, n_group=4, topk_group=2,
        first_k_dense_replace=1, max_position_embeddings=128)
    base.update(kw)
    return DeepseekV2Config(**base)


def test_mla_projection_shapes():
    cfg = tiny_cfg()
    attn = DeepseekV2Attention(cfg)
    # low-rank path present and sized
    assert attn.q_a_proj.out_features == 24
    assert attn.kv_a_proj_with_mqa.out_features == 16 + 8  # latent + rope
    assert attn.kv_b_proj.in_features == 16
    x = torch.randn(2, 10, 64)
    out = attn(x)
    assert out.shape == (2, 10, 64)

    # V2-Lite style: no q compression
    attn2 = DeepseekV2Attention(tiny_cfg(q_lora_rank=None))
    assert hasattr(attn2, "q_proj") and not hasattr(attn2, "q_a_proj")
    assert attn2(x).shape == (2, 10, 64)


def test_group_limited_routing():
    cfg = tiny_cfg()  # 8 experts in 4 groups of 2, topk_group=2
    gate = DeepseekV2MoEGate(cfg)
    x = torch.randn(16, 64)
    topk_w, topk_e, logits = gate(x)
    assert topk_e.shape == (16, 2) and logits.shape == (16, 8)
    # every selected expert must live in one of the token's top-2 groups
    scores = logits.softmax(-1).view(16, 4, 2)
    group_scores = scores.amax(-1)
    top_groups = group_scores.topk(2, dim=-1).indices
    chosen_groups = topk_e // 2
    for t in range(16):
        assert set(chosen_groups[t].tolist()) <= set(top_groups[t].tolist())
    # routed_scaling_factor applied
    cfg2 = tiny_cfg(routed_scaling_factor=2.0)
    gate2 = DeepseekV2MoEGate(cfg2)
    gate2.weight.data.copy_(gate.weight.data)
    w2, _, _ = gate2(x)
    torch.testing.assert_close(w2, topk_w * 2.0)


def test_dense_and_moe_layer_layout():
    m = DeepseekV2ForCausalLM(tiny_cfg())
    layers = m.deepseek_v2.layers
    assert not layers[0].is_moe      # first_k_dense_replace=1
    assert layers[1].is_moe and layers[2].is_moe
    assert layers[1].mlp.shared_experts is not None


def test_forward_backward_and_cache_parity():
    m = DeepseekV2ForCausalLM(tiny_cfg())
    ids = torch.randint(0, 128, (2, 16))
    loss, logits = m(input_ids=ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
    # MoE gate gets gradients
    assert m.deepseek_v2.layers[1].mlp.gate.weight.grad is not None

    m.eval()
    with torch.no_grad():
        full = m(input_ids=ids)
        lg, past = m(input_ids=ids[:, :-1], use_cache=True)
        step, _ = m(input_ids=ids[:, -1:], use_cache=True, past_key_values=past)
    torch.testing.assert_close(step[:, 0], full[:, -1], rtol=1e-4, atol=1e-4)


def test_generate():
    m = DeepseekV2ForCausalLM(tiny_cfg()).eval()
    ids = torch.randint(0, 128, (2, 6))
    out, _ = m.generate(ids, max_new_tokens=4, do_sample=False)
    assert out.shape == (2, 4)
