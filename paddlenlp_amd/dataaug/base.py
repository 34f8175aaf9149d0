"""Data-augmentation base: token selection and count logic.

Reference behavior: paddlenlp/dataaug/base_augment.py (aug_n / aug_percent /
aug_min / aug_max selection; stop-word skipping).  Tokenization here is
whitespace-based with a pluggable tokenizer; augmentation sources are local
(random vocab / user dictionaries) since this environment has no downloads.
"""
from __future__ import annotations

import random
from typing import Callable, List, Optional


class BaseAugment:
    def __init__(self, create_n: int = 1, aug_n: Optional[int] = None,
                 aug_percent: float = 0.1, aug_min: int = 1, aug_max: int = 10,
                 stop_words: Optional[List[str]] = None,
                 tokenizer: Optional[Callable[[str], List[str]]] = None,
                 seed: Optional[int] = None):
        self.create_n = create_n
        self.aug_n = aug_n
        self.aug_percent = aug_percent
        self.aug_min = aug_min
        self.aug_max = aug_max
        self.stop_words = set(stop_words or [])
        self.tokenize = tokenizer or (lambda s: s.split())
        self.rng = random.Random(seed)

    def _get_aug_n(self, seq_len: int, candidates: int) -> int:
        if candidates == 0:
            return 0
        n = self.aug_n if self.aug_n is not None else int(seq_len * self.aug_percent)
        n = max(self.aug_min, min(self.aug_max, n))
        return min(n, candidates)

    def _aug_indexes(self, tokens: List[str]) -> List[int]:
        return [i for i, t in enumerate(tokens) if t not in self.stop_words]

    def augment(self, sequence):
        """str -> List[str] of create_n augmented variants (or a list of
        inputs -> list of lists, reference augment())."""
        if isinstance(sequence, (list, tuple)):
            return [self.augment(s) for s in sequence]
        return [self._augment_once(sequence) for _ in range(self.create_n)]

    def _augment_once(self, sequence: str) -> str:
        raise NotImplementedError
