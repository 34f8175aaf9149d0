"""Stopping criteria (reference: paddlenlp/generation StoppingCriteria).

Callable objects deciding when a decode loop must stop, composable via
StoppingCriteriaList (logical OR).
"""
from __future__ import annotations

import time
from typing import List, Optional

import torch


class StoppingCriteria:
    def __call__(self, input_ids: torch.Tensor, scores) -> bool:
        raise NotImplementedError


class MaxLengthCriteria(StoppingCriteria):
    def __init__(self, max_length: int):
        self.max_length = max_length

    def __call__(self, input_ids, scores) -> bool:
        return input_ids.shape[-1] >= self.max_length


class MaxNewTokensCriteria(StoppingCriteria):
    def __init__(self, start_length: int, max_new_tokens: int):
        self.start_length = start_length
        self.max_new_tokens = max_new_tokens

    def __call__(self, input_ids, scores) -> bool:
        return input_ids.shape[-1] >= self.start_length + self.max_new_tokens


class MaxTimeCriteria(StoppingCriteria):
    def __init__(self, max_time: float, initial_timestamp: Optional[float] = None):
        self.max_time = max_time
        self.initial_timestamp = (time.time() if initial_timestamp is None
                                  else initial_timestamp)

    def __call__(self, input_ids, scores) -> bool:
        return time.time() - self.initial_timestamp > self.max_time


class StopStringsCriteria(StoppingCriteria):
    """Stop when every sequence's decoded tail contains one of the strings."""

    def __init__(self, tokenizer, stop_strings: List[str], window: int = 16):
        self.tokenizer = tokenizer
        self.stop_strings = list(stop_strings)
        self.window = window

    def __call__(self, input_ids, scores) -> bool:
        for row in input_ids:
            tail = self.tokenizer.decode(row[-self.window:].tolist())
            if not any(ss in tail for ss in self.stop_strings):
                return False
        return True


class StoppingCriteriaList(list):
    def __call__(self, input_ids, scores) -> bool:
        return any(c(input_ids, scores) for c in self)
