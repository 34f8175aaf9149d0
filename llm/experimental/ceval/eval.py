"""C-Eval style multiple-choice evaluation (reference: llm/experimental/
ceval/eval.py + model_evaluator.py).

The reference scores each (question, option) continuation with the model
and picks the highest-likelihood option; accuracy per subject.  Same
protocol here over local JSON/JSONL files (the dataset itself is not
bundled — no network), for any framework CausalLM + tokenizer.
"""
from __future__ import annotations

import json
import os
from typing import Dict, List

import torch

CHOICES = ["A", "B", "C", "D"]


def load_ceval_split(path: str) -> List[dict]:
    """Rows: {"question": ..., "A"-"D": options, "answer": "A"}."""
    rows = []
    with open(path, encoding="utf-8") as f:
        if path.endswith(".jsonl"):
            rows = [json.loads(l) for l in f if l.strip()]
        else:
            rows = json.load(f)
    return rows


def _option_loglik(model, tokenizer, prompt: str, option: str,
                   device) -> float:
    """Sum log p(option tokens | prompt) under the causal LM."""
    p_ids = tokenizer(prompt)["input_ids"]
    full_ids = tokenizer(prompt + option)["input_ids"]
    if len(full_ids) <= len(p_ids):
        return float("-inf")
    ids = torch.tensor([full_ids], device=device)
    with torch.no_grad():
        logits = model(input_ids=ids)
        if isinstance(logits, tuple):
            logits = logits[0]
    logp = torch.log_softmax(logits.float(), dim=-1)
    total = 0.0
    for t in range(len(p_ids), len(full_ids)):
        total += float(logp[0, t - 1, full_ids[t]])
    return total


def evaluate_mcq(model, tokenizer, rows: List[dict],
                 subject: str = "") -> Dict[str, float]:
    """Likelihood-ranked multiple choice accuracy."""
    device = next(model.parameters()).device
    model.eval()
    correct = 0
    for row in rows:
        prompt = (f"以下是关于{subject}的单项选择题。\n" if subject else "") + \
            f"{row['question']}\n答案："
        scores = [_option_loglik(model, tokenizer, prompt, row[c], device)
                  for c in CHOICES if c in row]
        pred = CHOICES[max(range(len(scores)), key=scores.__getitem__)]
        correct += int(pred == row.get("answer"))
    n = max(1, len(rows))
    return {"accuracy": correct / n, "num": len(rows)}
