"""Character-level augmentation: substitute / insert / swap / delete.

Reference behavior: paddlenlp/dataaug/char.py (CharSubstitute :28,
CharInsert :247, CharSwap :449, CharDelete :515) — operates on characters
inside randomly chosen tokens.
"""
from __future__ import annotations

from typing import List, Optional

from .base import BaseAugment


class _CharAugment(BaseAugment):
    def _char_positions(self, token: str) -> List[int]:
        return list(range(len(token)))

    def _augment_once(self, sequence: str) -> str:
        tokens = self.tokenize(sequence)
        idxs = [i for i in self._aug_indexes(tokens) if len(tokens[i]) >= 2]
        n = self._get_aug_n(len(tokens), len(idxs))
        for i in self.rng.sample(idxs, n):
            tokens[i] = self._augment_token(tokens[i])
        return " ".join(tokens)

    def _augment_token(self, token: str) -> str:
        raise NotImplementedError


class CharSubstitute(_CharAugment):
    def __init__(self, alphabet: Optional[str] = None, **kwargs):
        super().__init__(**kwargs)
        self.alphabet = alphabet or "abcdefghijklmnopqrstuvwxyz"

    def _augment_token(self, token: str) -> str:
        pos = self.rng.randrange(len(token))
        ch = self.rng.choice([c for c in self.alphabet if c != token[pos]])
        return token[:pos] + ch + token[pos + 1:]


class CharInsert(_CharAugment):
    def __init__(self, alphabet: Optional[str] = None, **kwargs):
        super().__init__(**kwargs)
        self.alphabet = alphabet or "abcdefghijklmnopqrstuvwxyz"

    def _augment_token(self, token: str) -> str:
        pos = self.rng.randrange(len(token) + 1)
        return token[:pos] + self.rng.choice(self.alphabet) + token[pos:]


class CharSwap(_CharAugment):
    def _augment_token(self, token: str) -> str:
        pos = self.rng.randrange(len(token) - 1)
        return (token[:pos] + token[pos + 1] + token[pos] + token[pos + 2:])


class CharDelete(_CharAugment):
    def _augment_token(self, token: str) -> str:
        pos = self.rng.randrange(len(token))
        return token[:pos] + token[pos + 1:]
