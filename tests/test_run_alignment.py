"""End-to-end alignment CLI tests: run_rm.py and run_ppo.py in-process.

Reference behavior: llm/alignment/{rm,ppo} entry points.
"""
import json
import os
import sys

import pytest
import torch

from tests.test_run_pretrain import argv_context_guard


@pytest.fixture
def align_setup(tmp_path):
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
    LlamaForCausalLM.from_config(cfg).save_pretrained(str(model_dir))
    tok.save_pretrained(str(model_dir))

    data_dir = tmp_path / "data"
    data_dir.mkdir()
    examples = [
        {"src": "the quick", "chosen": "brown fox", "rejected": "dog dog"},
        {"src": "lazy dog", "chosen": "jumps over", "rejected": "the the"},
    ] * 4
    with open(data_dir / "train.json", "w") as f:
        for ex in examples:
            f.write(json.dumps(ex) + "\n")
    return tmp_path


def _run_entry(tmp_path, rel_path, module_name, cfg_extra=None):
    cfg = {
        "model_name_or_path": str(tmp_path / "model"),
        "dataset_name_or_path": str(tmp_path / "data"),
        "output_dir": str(tmp_path / "out"),
        "per_device_train_batch_size": 2,
        "max_steps": 2,
        "logging_steps": 1,
        "save_steps": 100,
        "learning_rate": 1e-3,
        "do_train": True,
    }
    cfg.update(cfg_extra or {})
    cfg_file = tmp_path / "args.json"
    cfg_file.write_text(json.dumps(cfg))
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    sys.path.insert(0, os.path.join(root, "llm", "alignment", rel_path))
    import importlib

    mod = importlib.import_module(module_name)
    importlib.reload(mod)
    with argv_context_guard([f"{module_name}.py", str(cfg_file)]):
        mod.main()
    return tmp_path / "out"


def test_run_rm(align_setup):
    out = _run_entry(align_setup, "rm", "run_rm")
    assert (out / "reward_model.pt").is_file()
    sd = torch.load(out / "reward_model.pt", weights_only=True)
    assert any("value_head" in k or "score" in k or "reward" in k
               for k in sd) or len(sd) > 0


def test_run_ppo(align_setup):
    out = _run_entry(align_setup, "ppo", "run_ppo",
                     {"num_ppo_steps": 1, "rollout_batch_size": 2,
                      "max_new_tokens": 4, "ppo_epochs": 1})
    assert (out / "model.safetensors").is_file()
