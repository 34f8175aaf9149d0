"""Model-infra tests: tiny Llama, save/load, Auto registry, configs."""
import os
import tempfile

import pytest
import torch

from paddlenlp_amd.transformers import (
    AutoConfig,
    AutoModelForCausalLM,
    LlamaConfig,
    LlamaForCausalLM,
)


@pytest.fixture
def tiny_config():
    return LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=64, dtype="float32",
    )


def test_forward_backward(tiny_config):
    model = LlamaForCausalLM.from_config(tiny_config)
    ids = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    loss, logits = model(input_ids=ids, labels=labels)
    assert logits.shape == (2, 16, 128)
    loss.backward()
    assert all(p.grad is not None for p in model.parameters())


def test_unfused_matches_fused(tiny_config):
    """fuse_attention_qkv / fuse_attention_ffn change layout, not math."""
    torch.manual_seed(0)
    fused = LlamaForCausalLM.from_config(tiny_config)
    cfg2 = LlamaConfig(**{**tiny_config.to_dict(),
                          "fuse_attention_qkv": False, "fuse_attention_ffn": False})
    unfused = LlamaForCausalLM.from_config(cfg2)
    # copy fused weights into the unfused layout
    sd = fused.state_dict()
    new_sd = {}
    d = tiny_config.head_dim
    qo = tiny_config.num_attention_heads * d
    kvo = tiny_config.num_key_value_heads * d
    for k, v in sd.items():
        if "qkv_proj" in k:
            q, kk, vv = v.split([qo, kvo, kvo], dim=0)
            new_sd[k.replace("qkv_proj", "q_proj")] = q
            new_sd[k.replace("qkv_proj", "k_proj")] = kk
            new_sd[k.replace("qkv_proj", "v_proj")] = vv
        elif "gate_up_fused_proj" in k:
            g, u = v.chunk(2, dim=0)
            new_sd[k.replace("gate_up_fused_proj", "gate_proj")] = g
            new_sd[k.replace("gate_up_fused_proj", "up_proj")] = u
        else:
            new_sd[k] = v
    unfused.load_state_dict(new_sd)
    ids = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        out1 = fused(input_ids=ids)
        out2 = unfused(input_ids=ids)
    assert torch.allclose(out1, out2, atol=1e-5)


def test_save_load_roundtrip(tiny_config):
    model = LlamaForCausalLM.from_config(tiny_config)
    ids = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        ref = model(input_ids=ids)
    with tempfile.TemporaryDirectory() as d:
        model.save_pretrained(d)
        assert os.path.isfile(os.path.join(d, "config.json"))
        m2 = LlamaForCausalLM.from_pretrained(d)
        with torch.no_grad():
            out = m2(input_ids=ids)
        assert torch.equal(ref, out)
        # auto registry resolves from config.json
        m3 = AutoModelForCausalLM.from_pretrained(d)
        assert isinstance(m3, LlamaForCausalLM)
        cfg = AutoConfig.from_pretrained(d)
        assert cfg.hidden_size == 64


def test_sharded_save(tiny_config):
    model = LlamaForCausalLM.from_config(tiny_config)
    with tempfile.TemporaryDirectory() as d:
        model.save_pretrained(d, max_shard_size=50_000)  # force sharding
        files = os.listdir(d)
        assert "model.safetensors.index.json" in files
        assert sum(f.endswith(".safetensors") for f in files) > 1
        m2 = LlamaForCausalLM.from_pretrained(d)
        for (n1, p1), (n2, p2) in zip(model.named_parameters(), m2.named_parameters()):
            assert torch.equal(p1, p2), n1


def test_kv_cache_decode(tiny_config):
    """Prefill + cached decode == full forward."""
    torch.manual_seed(0)
    model = LlamaForCausalLM.from_config(tiny_config)
    model.eval()
    ids = torch.randint(0, 128, (1, 12))
    with torch.no_grad():
        full = model(input_ids=ids)
        logits_p, cache = model(input_ids=ids[:, :-1], use_cache=True)
        logits_d, _ = model(input_ids=ids[:, -1:], past_key_values=cache, use_cache=True)
    assert torch.allclose(full[:, -1], logits_d[:, 0], atol=1e-4), \
        (full[:, -1] - logits_d[:, 0]).abs().max()


def test_attribute_map(tiny_config):
    assert tiny_config.n_embd == 64
    tiny_config.n_layer = 3
    assert tiny_config.num_hidden_layers == 3


def test_rope_scaling_variants():
    for stype in [None, "linear", "ntk", "llama3"]:
        cfg = LlamaConfig(
            vocab_size=64, hidden_size=32, intermediate_size=64,
            num_hidden_layers=1, num_attention_heads=2, num_key_value_heads=2,
            max_position_embeddings=32,
            rope_scaling_type=stype, rope_scaling_factor=2.0,
        )
        model = LlamaForCausalLM.from_config(cfg)
        out = model(input_ids=torch.randint(0, 64, (1, 8)))
        assert torch.isfinite(out).all(), stype
