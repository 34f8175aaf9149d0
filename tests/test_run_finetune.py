"""End-to-end SFT CLI tests: run llm/run_finetune.py in-process (full FT,
LoRA, ZeroPadding+FlashMask)."""
import json
import os
import sys

import pytest
import torch

from tests.test_run_pretrain import argv_context_guard


@pytest.fixture
def sft_setup(tmp_path):
    from tests.test_inference_engine import _make_tiny_tokenizer
    from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM

    tok = _make_tiny_tokenizer()
    torch.manual_seed(0)
    model_dir = tmp_path / "model"
    cfg = LlamaConfig(
        vocab_size=16, hidden_size=32, intermediate_size=64,
        num_hidden_layers=2, num_attention_heads=2, num_key_value_heads=2,
        max_position_embeddings=128, dtype="float32", eos_token_id=2,
    )
    model = LlamaForCausalLM.from_config(cfg)
    model.save_pretrained(str(model_dir))
    tok.save_pretrained(str(model_dir))

    data_dir = tmp_path / "data"
    data_dir.mkdir()
    examples = [
        {"src": "the quick brown", "tgt": "fox jumps"},
        {"src": "lazy dog and", "tgt": "the fox"},
        {"src": "a quick dog", "tgt": "jumps over"},
        {"src": "the lazy fox", "tgt": "and a dog"},
    ] * 4
    with open(data_dir / "train.json", "w") as f:
        for ex in examples:
            f.write(json.dumps(ex) + "\n")
    return tmp_path


def _run_finetune(tmp_path, extra=()):
    cfg = {
        "model_name_or_path": str(tmp_path / "model"),
        "dataset_name_or_path": str(tmp_path / "data"),
        "output_dir": str(tmp_path / "out"),
        "max_length": 32,
        "per_device_train_batch_size": 2,
        "max_steps": 4,
        "logging_steps": 2,
        "save_steps": 100,
        "learning_rate": 1e-3,
        "do_train": True,
    }
    cfg_file = tmp_path / "sft.json"
    cfg_file.write_text(json.dumps(cfg))
    sys.path.insert(0, os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))), "llm"))
    import importlib

    import run_finetune

    importlib.reload(run_finetune)
    with argv_context_guard(["run_finetune.py", str(cfg_file), *extra]):
        run_finetune.main()
    return tmp_path / "out"


def test_run_finetune_full(sft_setup):
    out = _run_finetune(sft_setup)
    assert (out / "model.safetensors").is_file()


def test_run_finetune_lora(sft_setup):
    out = _run_finetune(sft_setup, ("--lora", "true", "--lora_rank", "4"))
    assert (out / "lora_model_state.safetensors").is_file()
    assert (out / "lora_config.json").is_file()


def test_run_finetune_zero_padding_flashmask(sft_setup):
    out = _run_finetune(sft_setup, ("--zero_padding", "true", "--flash_mask", "true"))
    assert (out / "model.safetensors").is_file()


def test_run_finetune_qlora(sft_setup):
    out = _run_finetune(sft_setup, ("--lora", "true", "--lora_rank", "4",
                                    "--weight_quantize_algo", "nf4"))
    assert (out / "lora_model_state.safetensors").is_file()


def test_run_finetune_ptq_flow(sft_setup):
    """do_ptq calibrates + swaps linears + saves a quantized state dict
    (reference apply_ptq flow)."""
    out = _run_finetune(sft_setup,
                        extra=["--do_ptq", "1", "--ptq_step", "2",
                               "--act_quant_method", "avg"])
    assert (out / "ptq" / "quantized_model.pt").is_file()
