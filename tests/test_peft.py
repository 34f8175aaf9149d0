"""PEFT tests: LoRA wrap/train/save/merge, prefix tuning, finetune CLI."""
import contextlib
import json
import os
import sys

import pytest
import torch

from paddlenlp_amd.peft import LoRAConfig, LoRAModel, PrefixConfig, PrefixModelForCausalLM
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM


def tiny_llama(seed=0, **kw):
    torch.manual_seed(seed)
    cfg = LlamaConfig(
        vocab_size=128, hidden_size=64, intermediate_size=128,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=2,
        max_position_embeddings=128, dtype="float32", **kw,
    )
    return LlamaForCausalLM.from_config(cfg)


def test_lora_wrap_and_trainable():
    model = tiny_llama()
    lora = LoRAModel(model, LoRAConfig(r=4))
    trainable = [n for n, p in lora.named_parameters() if p.requires_grad]
    frozen = [n for n, p in lora.named_parameters() if not p.requires_grad]
    assert all("lora_" in n for n in trainable) and len(trainable) > 0
    assert any("embed_tokens" in n for n in frozen)
    ids = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    loss, _ = lora(input_ids=ids, labels=labels)
    loss.backward()
    for n, p in lora.named_parameters():
        if "lora_B" in n:
            assert p.grad is not None


def test_lora_zero_init_is_identity():
    """lora_B starts at 0 -> wrapped model output == base model output."""
    model = tiny_llama(seed=3)
    ids = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        ref = model(input_ids=ids)
    lora = LoRAModel(model, LoRAConfig(r=4))
    with torch.no_grad():
        out = lora(input_ids=ids)
    assert torch.allclose(out, ref, atol=1e-6)


def test_lora_merge_equals_adapter():
    model = tiny_llama(seed=4)
    lora = LoRAModel(model, LoRAConfig(r=4))
    # give lora_B non-zero values so the adapter actually does something
    with torch.no_grad():
        for n, p in lora.named_parameters():
            if "lora_B" in n:
                p.normal_(0, 0.02)
    ids = torch.randint(0, 128, (2, 16))
    with torch.no_grad():
        before = lora(input_ids=ids)
    lora.merge()
    with torch.no_grad():
        after = lora(input_ids=ids)
    assert torch.allclose(before, after, atol=1e-4), (before - after).abs().max()


def test_lora_save_load(tmp_path):
    model = tiny_llama(seed=5)
    lora = LoRAModel(model, LoRAConfig(r=4))
    with torch.no_grad():
        for n, p in lora.named_parameters():
            if "lora_B" in n:
                p.normal_(0, 0.02)
    ids = torch.randint(0, 128, (1, 8))
    with torch.no_grad():
        ref = lora(input_ids=ids)
    lora.save_pretrained(str(tmp_path))
    model2 = tiny_llama(seed=5)
    lora2 = LoRAModel.from_pretrained(model2, str(tmp_path))
    with torch.no_grad():
        out = lora2(input_ids=ids)
    assert torch.allclose(out, ref, atol=1e-6)


def test_prefix_tuning():
    model = tiny_llama(seed=6)
    prefix = PrefixModelForCausalLM(model, PrefixConfig(num_prefix_tokens=4))
    trainable = [n for n, p in prefix.named_parameters() if p.requires_grad]
    assert all("prefix_encoder" in n for n in trainable) and trainable
    ids = torch.randint(0, 128, (2, 16))
    labels = torch.randint(0, 128, (2, 16))
    loss, _ = prefix(input_ids=ids, labels=labels)
    loss.backward()
    assert torch.isfinite(loss)
    assert prefix.prefix_encoder.embedding.weight.grad is not None


def test_sft_convert_example():
    sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "llm"))
    from utils.data import convert_example

    class FakeTok:
        eos_token_id = 2

        def encode(self, text):
            return [ord(c) % 100 for c in text]

    ex = {"src": "abc", "tgt": "de"}
    out = convert_example(ex, FakeTok(), max_length=32)
    assert len(out["input_ids"]) == len(out["labels"])
    # prompt region masked
    assert out["labels"][0] == -100 and out["labels"][1] == -100
    # response region supervised
    assert out["labels"][-1] == 2  # eos supervised
