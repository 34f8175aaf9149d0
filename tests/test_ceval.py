"""C-Eval MCQ harness (reference llm/experimental/ceval)."""
import json

import torch

from llm.experimental.ceval import evaluate_mcq, load_ceval_split
from paddlenlp_amd.transformers import LlamaConfig, LlamaForCausalLM


class VocabTok:
    def __call__(self, text):
        return {"input_ids": [ord(c) % 96 for c in text]}


def test_mcq_likelihood_ranking(tmp_path):
    rows = [
        {"question": "q1", "A": "aa", "B": "bb", "C": "cc", "D": "dd",
         "answer": "B"},
        {"question": "q2", "A": "xy", "B": "yz", "C": "zx", "D": "xx",
         "answer": "A"},
    ]
    f = tmp_path / "dev.json"
    f.write_text(json.dumps(rows), encoding="utf-8")
    loaded = load_ceval_split(str(f))
    assert loaded == rows

    torch.manual_seed(0)
    cfg = LlamaConfig(vocab_size=96, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=2, max_position_embeddings=64)
    model = LlamaForCausalLM.from_config(cfg)
    out = evaluate_mcq(model, VocabTok(), loaded, subject="test")
    assert out["num"] == 2 and 0.0 <= out["accuracy"] <= 1.0
