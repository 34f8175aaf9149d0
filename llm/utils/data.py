"""SFT example tokenization (reference: llm/utils/data.py:72-249).

Input examples use the reference's format: {"src": prompt, "tgt": response}
(or chat "messages").  Labels mask the prompt region with -100.
"""
from __future__ import annotations

from typing import Dict, List


def convert_example(example: Dict, tokenizer, max_length: int = 2048,
                    src_length: int = None, eos_token_id=None) -> Dict:
    if "messages" in example:
        prompt_text = tokenizer.apply_chat_template(
            example["messages"][:-1], tokenize=False, add_generation_prompt=True)
        full_text = prompt_text + example["messages"][-1]["content"]
    else:
        src = example.get("src") or example.get("instruction") or example.get("prompt")
        tgt = example.get("tgt") or example.get("output") or example.get("response")
        if isinstance(src, list):
            src = src[0]
        if isinstance(tgt, list):
            tgt = tgt[0]
        prompt_text = str(src)
        full_text = prompt_text + str(tgt)

    prompt_ids = tokenizer.encode(prompt_text)
    full_ids = tokenizer.encode(full_text)
    if src_length is not None and len(prompt_ids) > src_length:
        # truncate the prompt head, keep its tail + the full target
        # (reference src_length budgeting)
        drop = len(prompt_ids) - src_length
        prompt_ids = prompt_ids[drop:]
        full_ids = full_ids[drop:]
    eos = eos_token_id if eos_token_id is not None else tokenizer.eos_token_id
    if eos is not None:
        full_ids = full_ids + [eos]
    full_ids = full_ids[:max_length]
    n_prompt = min(len(prompt_ids), len(full_ids))

    input_ids = full_ids[:-1]
    labels = list(full_ids[1:])
    for i in range(min(n_prompt - 1, len(labels))):
        labels[i] = -100
    return {"input_ids": input_ids, "labels": labels}
