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


class NoRepeatNGramLogitsProcessor:
    """Ban tokens completing any already-seen n-gram (reference :177)."""

    def __init__(self, ngram_size: int):
        self.n = ngram_size

    def __call__(self, input_ids, logits):
        n = self.n
        if input_ids.shape[1] + 1 < n:
            return logits
        for b in range(input_ids.shape[0]):
            seq = input_ids[b].tolist()
            prefix = tuple(seq[-(n - 1):]) if n > 1 else ()
            banned = set()
            for i in range(len(seq) - n + 1):
                if tuple(seq[i:i + n - 1]) == prefix:
                    banned.add(seq[i + n - 1])
            if banned:
                logits[b, list(banned)] = -float("inf")
        return logits


class ForcedBOSTokenLogitsProcessor:
    """Force the first generated token (reference :252)."""

    def __init__(self, prompt_len: int, bos_token_id: int):
        self.prompt_len = prompt_len
        self.bos = bos_token_id

    def __call__(self, input_ids, logits):
        if input_ids.shape[1] == self.prompt_len:
            logits[:] = -float("inf")
            logits[:, self.bos] = 0.0
        return logits


class ForcedEOSTokenLogitsProcessor:
    """Force EOS at max length (reference :272)."""

    def __init__(self, max_total_len: int, eos_token_id: int):
        self.max_total_len = max_total_len
        self.eos = eos_token_id

    def __call__(self, input_ids, logits):
        if input_ids.shape[1] == self.max_total_len - 1:
            logits[:] = -float("inf")
            logits[:, self.eos] = 0.0
        return logits


class SequenceBiasLogitsProcessor:
    """Additive bias on single tokens or when a multi-token sequence's
    prefix matches the tail of input_ids (reference :375)."""

    def __init__(self, sequence_bias):
        # {tuple(token_ids): bias}
        self.bias = {tuple(k): float(v) for k, v in sequence_bias.items()}

    def __call__(self, input_ids, logits):
        for seq, bias in self.bias.items():
            if len(seq) == 1:
                logits[:, seq[0]] += bias
                continue
            pre, last = seq[:-1], seq[-1]
            L = len(pre)
            if input_ids.shape[1] < L:
                continue
            tail = input_ids[:, -L:]
            match = (tail == torch.tensor(
                pre, device=input_ids.device)).all(dim=-1)
            logits[match, last] += bias
        return logits


class NoBadWordsLogitsProcessor(SequenceBiasLogitsProcessor):
    """Hard-ban token sequences (reference :527 — -inf sequence bias)."""

    def __init__(self, bad_words_ids):
        super().__init__({tuple(w): -float("inf") for w in bad_words_ids})


class PrefixConstrainedLogitsProcessor:
    """Constrain each step to caller-approved tokens (reference :623)."""

    def __init__(self, prefix_allowed_tokens_fn, num_beams: int = 1):
        self.fn = prefix_allowed_tokens_fn
        self.num_beams = num_beams

    def __call__(self, input_ids, logits):
        mask = torch.full_like(logits, -float("inf"))
        for i in range(input_ids.shape[0]):
            allowed = self.fn(i // self.num_beams, input_ids[i])
            mask[i, list(allowed)] = 0.0
        return logits + mask
