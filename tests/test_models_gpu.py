"""GPU forward+backward smokes for the newer model families (encoder core,
T5, Qwen2-MoE, QLoRA) — verifies the HIP kernel seams (non-causal flash,
rms_norm, swiglu) under bf16 on gfx950.
"""
import pytest
import torch

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module")
def device():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    return torch.device("cuda:0")


def test_bert_mlm_gpu(device):
    from paddlenlp_amd.transformers import BertConfig, BertForMaskedLM

    torch.manual_seed(0)
    cfg = BertConfig(vocab_size=500, hidden_size=128, num_hidden_layers=2,
                     num_attention_heads=4, intermediate_size=256,
                     max_position_embeddings=128, hidden_dropout_prob=0.0,
                     attention_probs_dropout_prob=0.0)
    m = BertForMaskedLM(cfg).to(device).to(torch.bfloat16)
    ids = torch.randint(0, 500, (2, 64), device=device)
    labels = ids.clone()
    labels[:, :32] = -100
    loss, logits = m(ids, labels=labels)
    loss.backward()
    assert torch.isfinite(loss)
    assert m.bert.embeddings.word_embeddings.weight.grad is not None

    # non-causal flash kernel path vs SDPA-with-mask fallback parity
    m.zero_grad(set_to_none=True)
    with torch.no_grad():
        out_kernel = m(ids)                       # no mask -> HIP flash
        mask = torch.ones(2, 64, device=device)
        out_masked = m(ids, attention_mask=mask)  # mask -> SDPA fallback
    rel = (out_kernel - out_masked).abs().max() / out_masked.abs().max()
    assert rel < 0.05, rel


def test_t5_gpu(device):
    from paddlenlp_amd.transformers import T5Config, T5ForConditionalGeneration

    torch.manual_seed(0)
    cfg = T5Config(vocab_size=500, d_model=128, d_kv=32, d_ff=256,
                   num_layers=2, num_heads=4, dropout_rate=0.0)
    m = T5ForConditionalGeneration(cfg).to(device).to(torch.bfloat16)
    src = torch.randint(0, 500, (2, 32), device=device)
    labels = torch.randint(1, 500, (2, 16), device=device)
    loss, _ = m(input_ids=src, labels=labels)
    loss.backward()
    assert torch.isfinite(loss)
    out, _ = m.generate(src, max_new_tokens=4, do_sample=False)
    assert out.shape[0] == 2


def test_qwen2_moe_gpu(device):
    from paddlenlp_amd.transformers import Qwen2MoeConfig, Qwen2MoeForCausalLM

    torch.manual_seed(0)
    cfg = Qwen2MoeConfig(vocab_size=500, hidden_size=128, intermediate_size=256,
                         num_hidden_layers=2, num_attention_heads=4,
                         num_key_value_heads=2, num_experts=4,
                         num_experts_per_tok=2, moe_intermediate_size=64,
                         shared_expert_intermediate_size=128,
                         max_position_embeddings=128)
    m = Qwen2MoeForCausalLM(cfg).to(device).to(torch.bfloat16)
    ids = torch.randint(0, 500, (2, 64), device=device)
    loss, _ = m(input_ids=ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)


def test_qlora_nf4_gpu(device):
    from paddlenlp_amd.peft import LoRAConfig, LoRAModel
    from paddlenlp_amd.quantization import (
        QuantizationConfig, replace_with_quantization_linear)
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=500, hidden_size=128, intermediate_size=256,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=128)
    m = LlamaForCausalLM(cfg).to(device).to(torch.bfloat16)
    replace_with_quantization_linear(
        m, QuantizationConfig(weight_quantize_algo="nf4"))
    lora = LoRAModel(m, LoRAConfig(r=4))
    ids = torch.randint(0, 500, (2, 32), device=device)
    loss, _ = lora.model(input_ids=ids, labels=ids)
    loss.backward()
    b_grads = [p.grad for n, p in lora.named_parameters()
               if p.requires_grad and "lora_B" in n]
    assert b_grads and all(g is not None and torch.isfinite(g).all()
                           for g in b_grads)


def test_deepseek_v2_gpu(device):
    from paddlenlp_amd.transformers import DeepseekV2Config, DeepseekV2ForCausalLM

    torch.manual_seed(0)
    cfg = DeepseekV2Config(
        vocab_size=512, hidden_size=128, intermediate_size=256,
        moe_intermediate_size=64, num_hidden_layers=2, num_attention_heads=4,
        q_lora_rank=32, kv_lora_rank=16, qk_nope_head_dim=16,
        qk_rope_head_dim=8, v_head_dim=16, n_routed_experts=4,
        n_shared_experts=1, num_experts_per_tok=2, n_group=2, topk_group=1,
        first_k_dense_replace=1, max_position_embeddings=128)
    m = DeepseekV2ForCausalLM(cfg).to(device).to(torch.bfloat16)
    ids = torch.randint(3, 512, (2, 32), device=device)
    loss, _ = m(input_ids=ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)


def test_jamba_gpu(device):
    from paddlenlp_amd.transformers import JambaConfig, JambaForCausalLM

    torch.manual_seed(0)
    cfg = JambaConfig(vocab_size=512, hidden_size=128, intermediate_size=256,
                      num_hidden_layers=4, num_attention_heads=4,
                      num_key_value_heads=2, attn_layer_period=4,
                      attn_layer_offset=1, expert_layer_period=2,
                      expert_layer_offset=0, num_experts=4,
                      num_experts_per_tok=2, mamba_d_state=8,
                      max_position_embeddings=128)
    m = JambaForCausalLM(cfg).to(device).to(torch.bfloat16)
    ids = torch.randint(3, 512, (2, 32), device=device)
    loss, _ = m(input_ids=ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)
