"""LLM serving utilities.

Reference behavior: paddlenlp/utils/llm_utils.py — `read_res` :753 (the
detokenizer worker that streams generated token ids out of a queue and
decodes them off the GPU path) and `get_rotary_position_embedding` :784.
"""
from __future__ import annotations

from typing import Optional

import torch

from .log import logger


def get_rotary_position_embedding(position_ids: torch.Tensor, head_dim: int,
                                  rope_theta: float = 10000.0,
                                  rope_scaling: Optional[dict] = None):
    """position_ids [1, S] -> rotary embedding [2, 1, S, 1, head_dim//2]
    (cos stacked over sin), with optional linear/ntk scaling."""
    base = rope_theta
    factor = 1.0
    if rope_scaling:
        scaling_type = rope_scaling.get("type", rope_scaling.get("rope_type"))
        scaling_factor = float(rope_scaling.get("factor", 1.0))
        if scaling_type == "linear":
            factor = scaling_factor
        elif scaling_type in ("ntk", "dynamic_ntk", "dynamic"):
            base = rope_theta * scaling_factor ** (head_dim / (head_dim - 2))
        else:
            logger.warning(f"unknown rope_scaling type {scaling_type!r}: ignored")
    inv_freq = 1.0 / (base ** (
        torch.arange(0, head_dim, 2, dtype=torch.float32,
                     device=position_ids.device) / head_dim))
    pos = position_ids[0].float() / factor
    freqs = torch.outer(pos, inv_freq)                     # [S, D/2]
    out = torch.stack([freqs.cos(), freqs.sin()])          # [2, S, D/2]
    return out[:, None, :, None, :]                        # [2, 1, S, 1, D/2]


def read_res(tokenizer, result_queue, output_queue, eos_token_id=None,
             timeout: float = 30.0):
    """Detokenizer worker loop: pull per-step token-id lists for a batch of
    sequences from result_queue, accumulate, and push decoded strings to
    output_queue when a sequence finishes (id == eos or `None` sentinel ends
    the stream).  Runs on CPU so decoding never blocks GPU decode steps."""
    eos = eos_token_id if eos_token_id is not None else getattr(
        tokenizer, "eos_token_id", None)
    buffers = {}
    finished = set()
    while True:
        item = result_queue.get(timeout=timeout)
        if item is None:
            break
        for seq_id, token in item:
            if seq_id in finished:
                continue
            if eos is not None and token == eos:
                finished.add(seq_id)
                output_queue.put(
                    (seq_id, tokenizer.decode(buffers.get(seq_id, []),
                                              skip_special_tokens=True)))
                continue
            buffers.setdefault(seq_id, []).append(token)
    for seq_id, toks in buffers.items():
        if seq_id not in finished:
            output_queue.put(
                (seq_id, tokenizer.decode(toks, skip_special_tokens=True)))
