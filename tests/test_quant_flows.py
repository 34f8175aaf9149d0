"""LLM quantization flows: shift, smooth/AWQ, autoclip, PTQ, GPTQ.

Reference behavior: llm/utils/quant.py over paddleslim; here self-contained
torch (see llm/utils/quant.py docstring).
"""
import sys

import pytest
import torch

sys.path.insert(0, "llm")
from utils.quant import (  # noqa: E402
    apply_autoclip,
    apply_gptq,
    apply_ptq,
    apply_shift,
    apply_smooth,
    collect_activation_stats,
    gptq_quantize_weight,
    quantize_to_weight_only,
)

from paddlenlp_amd.quantization import QuantizationLinear
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

torch.manual_seed(0)


def tiny_llama():
    cfg = LlamaConfig(vocab_size=128, hidden_size=64, intermediate_size=128,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=128,
                      dtype="float32")
    return LlamaForCausalLM.from_config(cfg).eval()


def batches(n=4, B=2, S=16, seed=0):
    g = torch.Generator().manual_seed(seed)
    return [{"input_ids": torch.randint(0, 128, (B, S), generator=g)}
            for _ in range(n)]


def test_collect_stats():
    m = tiny_llama()
    stats = collect_activation_stats(m, iter(batches()), 4)
    key = "llama.layers.0.self_attn.qkv_proj"
    assert key in stats
    assert stats[key]["absmax"].shape == (64,)
    assert (stats[key]["absmax"] >= 0).all()


def test_shift_preserves_outputs():
    m = tiny_llama()
    ids = batches(1)[0]["input_ids"]
    with torch.no_grad():
        ref = m(input_ids=ids)
    apply_shift(m, iter(batches()), 4)
    # qkv gained a compensating bias; norm gained a shift buffer
    assert m.llama.layers[0].self_attn.qkv_proj.bias is not None
    assert hasattr(m.llama.layers[0].input_layernorm, "shift_bias")
    with torch.no_grad():
        out = m(input_ids=ids)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)


def test_smooth_preserves_outputs_and_flattens_activations():
    m = tiny_llama()
    ids = batches(1)[0]["input_ids"]
    with torch.no_grad():
        ref = m(input_ids=ids)
    before = collect_activation_stats(m, iter(batches()), 4)
    apply_smooth(m, iter(batches()), alpha=0.5)
    with torch.no_grad():
        out = m(input_ids=ids)
    torch.testing.assert_close(out, ref, rtol=1e-3, atol=1e-3)
    after = collect_activation_stats(m, iter(batches()), 4)
    key = "llama.layers.0.self_attn.qkv_proj"
    # dynamic range of the activation channels must compress
    rng = lambda s: float(s[key]["absmax"].max() / s[key]["absmax"].min().clamp(min=1e-8))
    assert rng(after) < rng(before)


def test_smooth_awq_grid():
    m = tiny_llama()
    ids = batches(1)[0]["input_ids"]
    with torch.no_grad():
        ref = m(input_ids=ids)
    apply_smooth(m, iter(batches()), do_awq=True, awq_grid=3)
    with torch.no_grad():
        out = m(input_ids=ids)
    torch.testing.assert_close(out, ref, rtol=1e-3, atol=1e-3)


def test_autoclip_bounds_error():
    m = tiny_llama()
    ids = batches(1)[0]["input_ids"]
    with torch.no_grad():
        ref = m(input_ids=ids)
    apply_autoclip(m, iter(batches()), 2, n_grid=5)
    with torch.no_grad():
        out = m(input_ids=ids)
    rel = (out - ref).abs().mean() / ref.abs().mean()
    assert rel < 0.2  # clipping trims outliers but keeps the model sane


def test_gptq_beats_round_to_nearest():
    torch.manual_seed(1)
    X = torch.randn(256, 32)
    # correlated channels make compensation matter
    X[:, 1] = X[:, 0] * 0.9 + 0.1 * X[:, 1]
    w = torch.randn(16, 32)
    H = X.t() @ X
    wq_gptq = gptq_quantize_weight(w, H, bits=4)
    qmax = 7
    scale = w.abs().amax(1, keepdim=True) / qmax
    wq_rtn = torch.clamp(torch.round(w / scale), -qmax, qmax) * scale
    err_gptq = ((X @ wq_gptq.t()) - (X @ w.t())).pow(2).mean()
    err_rtn = ((X @ wq_rtn.t()) - (X @ w.t())).pow(2).mean()
    assert err_gptq < err_rtn, (float(err_gptq), float(err_rtn))


def test_apply_gptq_model_level():
    m = tiny_llama()
    ids = batches(1)[0]["input_ids"]
    with torch.no_grad():
        ref = m(input_ids=ids)
    apply_gptq(m, iter(batches()), bits=8, num_batches=4)
    with torch.no_grad():
        out = m(input_ids=ids)
    rel = (out - ref).abs().mean() / ref.abs().mean()
    assert rel < 0.1, rel  # int8 gptq stays close
    # lm_head untouched
    assert isinstance(m.lm_head, torch.nn.Linear)


def test_ptq_and_pack():
    m = tiny_llama()
    apply_ptq(m, iter(batches()), algo="avg", num_batches=2)
    m2 = tiny_llama()
    quantize_to_weight_only(m2, "weight_only_int8")
    assert any(isinstance(mod, QuantizationLinear) for mod in m2.modules())
    ids = batches(1)[0]["input_ids"]
    with torch.no_grad():
        out = m2(input_ids=ids)
    assert torch.isfinite(out).all()


def test_shift_skips_norms_without_shift_support():
    """A norm class that does not apply shift_bias must be left alone
    (silent output corruption guard)."""
    from paddlenlp_amd.transformers import GemmaConfig, GemmaForCausalLM

    torch.manual_seed(0)
    cfg = GemmaConfig(vocab_size=96, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=4,
                      num_key_value_heads=2, head_dim=8,
                      max_position_embeddings=64)
    m = GemmaForCausalLM(cfg).eval()
    ids = torch.randint(3, 96, (1, 8))
    with torch.no_grad():
        ref = m(input_ids=ids)
    apply_shift(m, iter([{"input_ids": ids}]), 1)
    assert not hasattr(m.gemma.layers[0].input_layernorm, "shift_bias")
    with torch.no_grad():
        out = m(input_ids=ids)
    torch.testing.assert_close(out, ref)


def test_smooth_skips_non_linear_norms():
    from paddlenlp_amd.transformers import GemmaConfig, GemmaForCausalLM

    torch.manual_seed(0)
    cfg = GemmaConfig(vocab_size=96, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=1, num_attention_heads=4,
                      num_key_value_heads=2, head_dim=8,
                      max_position_embeddings=64)
    m = GemmaForCausalLM(cfg).eval()
    ids = torch.randint(3, 96, (1, 8))
    with torch.no_grad():
        ref = m(input_ids=ids)
    apply_smooth(m, iter([{"input_ids": ids}]), num_batches=1)
    with torch.no_grad():
        out = m(input_ids=ids)
    torch.testing.assert_close(out, ref)  # untouched
