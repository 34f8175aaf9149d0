"""Logits processors / warpers (reference: paddlenlp/generation/logits_process.py)."""
from __future__ import annotations

from typing import List

import torch


class LogitsProcessorList(list):
    def __call__(self, input_ids: torch.Tensor, logits: torch.Tensor) -> torch.Tensor:
        for proc in self:
            logits = proc(input_ids, logits)
        return logits


class RepetitionPenaltyLogitsProcessor:
    def __init__(self, penalty: float):
        self.penalty = penalty

    def __call__(self, input_ids, logits):
        if self.penalty == 1.0:
            return logits
        score = torch.gather(logits, 1, input_ids)
        score = torch.where(score < 0, score * self.penalty, score / self.penalty)
        logits.scatter_(1, input_ids, score)
        return logits


class MinNewTokensLengthLogitsProcessor:
    def __init__(self, prompt_len: int, min_new_tokens: int, eos_token_ids: List[int]):
        self.prompt_len = prompt_len
        self.min_new_tokens = min_new_tokens
        self.eos_token_ids = eos_token_ids

    def __call__(self, input_ids, logits):
        if input_ids.shape[-1] - self.prompt_len < self.min_new_tokens:
            for eos in self.eos_token_ids:
                logits[:, eos] = float("-inf")
        return logits


class TemperatureLogitsWarper:
    def __init__(self, temperature: float):
        self.temperature = max(temperature, 1e-6)

    def __call__(self, input_ids, logits):
        return logits / self.temperature


class TopKLogitsWarper:
    def __init__(self, top_k: int):
        self.top_k = top_k

    def __call__(self, input_ids, logits):
        if self.top_k <= 0 or self.top_k >= logits.shape[-1]:
            return logits
        kth = torch.topk(logits, self.top_k)[0][..., -1, None]
        return logits.masked_fill(logits < kth, float("-inf"))


class TopPLogitsWarper:
    def __init__(self, top_p: float):
        self.top_p = top_p

    def __call__(self, input_ids, logits):
        if self.top_p >= 1.0:
            return logits
        sorted_logits, sorted_idx = torch.sort(logits, descending=True)
        probs = sorted_logits.softmax(-1)
        cum = probs.cumsum(-1)
        remove = cum - probs > self.top_p  # keep first token exceeding p
        mask = remove.scatter(1, sorted_idx, remove)
        return logits.masked_fill(mask, float("-inf"))


class HammingDiversityLogitsProcessor:
    """Subtract `diversity_penalty` for tokens earlier beam groups already
    chose at this step (reference logits_process group beam search)."""

    def __init__(self, diversity_penalty: float, num_beams: int, num_beam_groups: int):
        self.diversity_penalty = diversity_penalty
        self.num_sub_beams = num_beams // num_beam_groups

    def __call__(self, scores, used_token_counts):
        """scores: [B*Kg, V]; used_token_counts: [B, V] picks by earlier
        groups this step."""
        if self.diversity_penalty == 0.0:
            return scores
        return scores - self.diversity_penalty * used_token_counts.repeat_interleave(
            self.num_sub_beams, dim=0)
