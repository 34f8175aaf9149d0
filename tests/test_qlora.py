"""QLoRA: 4-bit blockwise quantization (nf4/fp4, double-quant) + LoRA over a
quantized base.

Reference behavior: paddlenlp/quantization/qlora.py (blockwise quant/dequant),
quantization_utils.py:38 (replace_with_quantization_linear), peft/lora over
QuantizationLinear.
"""
import torch

from paddlenlp_amd.peft import LoRAConfig, LoRAModel
from paddlenlp_amd.quantization import (
    QuantizationConfig,
    QuantizationLinear,
    qlora_weight_dequantize,
    qlora_weight_quantize,
    qlora_weight_quantize_dequantize,
    replace_with_quantization_linear,
)
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

torch.manual_seed(0)


def tiny_llama(**kw):
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, **kw)
    return LlamaForCausalLM(cfg)


def test_nf4_roundtrip_accuracy():
    w = torch.randn(64, 128)
    for algo in ("nf4", "fp4"):
        for double_quant in (False, True):
            packed, state = qlora_weight_quantize(w, algo, double_quant=double_quant)
            assert packed.dtype == torch.uint8 and packed.numel() == w.numel() // 2
            wd = qlora_weight_dequantize(packed, state)
            rel = (w - wd).abs().mean() / w.abs().mean()
            assert rel < 0.15, (algo, double_quant, rel)


def test_double_quant_compresses_scales():
    w = torch.randn(256, 256)
    _, st = qlora_weight_quantize(w, "nf4", double_quant=True)
    # absmax stored int8 + fp32 super-scales instead of fp32 per block
    assert st["qabsmax"].dtype == torch.int8
    assert st["absmax_scale"].numel() == st["qabsmax"].numel() // 256


def test_quantize_dequantize_helper_matches():
    w = torch.randn(32, 64)
    wd = qlora_weight_quantize_dequantize(w, "nf4")
    packed, state = qlora_weight_quantize(w, "nf4")
    torch.testing.assert_close(wd, qlora_weight_dequantize(packed, state, w.dtype))


def test_replace_with_quantization_linear_skips_lm_head():
    model = tiny_llama()
    cfg = QuantizationConfig(weight_quantize_algo="nf4")
    replaced = replace_with_quantization_linear(model, cfg)
    assert replaced and all("lm_head" not in n for n in replaced)
    assert isinstance(model.llama.layers[0].self_attn.qkv_proj, QuantizationLinear)
    assert isinstance(model.lm_head, torch.nn.Linear)
    ids = torch.randint(0, 128, (2, 16))
    loss, _ = model(input_ids=ids, labels=ids)
    assert torch.isfinite(loss)


def test_qlora_trains_only_adapters():
    model = tiny_llama()
    ids = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        ref_loss, _ = model(input_ids=ids, labels=ids)

    replace_with_quantization_linear(
        model, QuantizationConfig(weight_quantize_algo="nf4"))
    lora = LoRAModel(model, LoRAConfig(r=4))
    trainable = [n for n, p in lora.named_parameters() if p.requires_grad]
    assert trainable and all("lora_" in n for n in trainable)

    # zero-init B => quantized forward unchanged by the adapters
    loss, _ = lora.model(input_ids=ids, labels=ids)
    # 4-bit base: loss differs from fp32 but stays finite and close-ish
    assert torch.isfinite(loss)

    loss.backward()
    named = [(n, p) for n, p in lora.named_parameters() if p.requires_grad]
    assert all(p.grad is not None for _, p in named)
    # with B zero-init, dA is zero on step 1 but dB must be non-zero
    b_grads = [p.grad for n, p in named if "lora_B" in n]
    assert b_grads and all(g.abs().sum() > 0 for g in b_grads)

    opt = torch.optim.AdamW([p for p in lora.parameters() if p.requires_grad], lr=1e-2)
    for _ in range(5):
        opt.zero_grad(set_to_none=True)
        loss, _ = lora.model(input_ids=ids, labels=ids)
        loss.backward()
        opt.step()
    final, _ = lora.model(input_ids=ids, labels=ids)
    assert float(final) < float(loss) + 1e-3  # adapters actually learn


def test_qlora_merge_refuses_packed_base():
    model = tiny_llama()
    replace_with_quantization_linear(
        model, QuantizationConfig(weight_quantize_algo="nf4"))
    lora = LoRAModel(model, LoRAConfig(r=4))
    layer = model.llama.layers[0].self_attn.qkv_proj
    try:
        layer.merge()
        raised = False
    except RuntimeError:
        raised = True
    assert raised
