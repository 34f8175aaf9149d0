"""Word-level augmentation: substitute / insert / swap / delete.

Reference behavior: paddlenlp/dataaug/word.py (WordSubstitute :29,
WordInsert, WordSwap, WordDelete).  Substitution sources: a user-provided
synonym dictionary ({word: [replacements]}) or a vocab list sampled at
random ("random" mode) — the reference's embedding/WordNet sources need
downloads, which this environment doesn't have.
"""
from __future__ import annotations

from typing import Dict, List, Optional

from .base import BaseAugment


class WordSubstitute(BaseAugment):
    def __init__(self, aug_type: str = "custom",
                 custom_dict: Optional[Dict[str, List[str]]] = None,
                 vocab: Optional[List[str]] = None, **kwargs):
        super().__init__(**kwargs)
        self.aug_type = aug_type
        self.custom_dict = custom_dict or {}
        self.vocab = vocab or []
        if aug_type == "random":
            assert self.vocab, "random substitution needs a vocab list"

    def _candidates(self, token: str) -> List[str]:
        if self.aug_type == "random":
            return [w for w in self.vocab if w != token]
        return [w for w in self.custom_dict.get(token, []) if w != token]

    def _augment_once(self, sequence: str) -> str:
        tokens = self.tokenize(sequence)
        idxs = [i for i in self._aug_indexes(tokens) if self._candidates(tokens[i])]
        n = self._get_aug_n(len(tokens), len(idxs))
        for i in self.rng.sample(idxs, n):
            tokens[i] = self.rng.choice(self._candidates(tokens[i]))
        return " ".join(tokens)


class WordInsert(BaseAugment):
    """Insert words (from custom dict keyed by the neighbor, or vocab)."""

    def __init__(self, aug_type: str = "random",
                 custom_dict: Optional[Dict[str, List[str]]] = None,
                 vocab: Optional[List[str]] = None, **kwargs):
        super().__init__(**kwargs)
        self.aug_type = aug_type
        self.custom_dict = custom_dict or {}
        self.vocab = vocab or []

    def _insert_for(self, token: str) -> Optional[str]:
        if self.aug_type == "custom":
            cands = self.custom_dict.get(token, [])
            return self.rng.choice(cands) if cands else None
        return self.rng.choice(self.vocab) if self.vocab else None

    def _augment_once(self, sequence: str) -> str:
        tokens = self.tokenize(sequence)
        idxs = self._aug_indexes(tokens)
        n = self._get_aug_n(len(tokens), len(idxs))
        for i in sorted(self.rng.sample(idxs, n), reverse=True):
            ins = self._insert_for(tokens[i])
            if ins is not None:
                tokens.insert(i + 1, ins)
        return " ".join(tokens)


class WordSwap(BaseAugment):
    """Swap adjacent words."""

    def _augment_once(self, sequence: str) -> str:
        tokens = self.tokenize(sequence)
        idxs = [i for i in self._aug_indexes(tokens) if i + 1 < len(tokens)]
        n = self._get_aug_n(len(tokens), len(idxs))
        for i in self.rng.sample(idxs, n):
            tokens[i], tokens[i + 1] = tokens[i + 1], tokens[i]
        return " ".join(tokens)


class WordDelete(BaseAugment):
    def _augment_once(self, sequence: str) -> str:
        tokens = self.tokenize(sequence)
        idxs = self._aug_indexes(tokens)
        n = self._get_aug_n(len(tokens), len(idxs))
        if n >= len(tokens):  # never delete everything
            n = max(0, len(tokens) - 1)
        drop = set(self.rng.sample(idxs, min(n, len(idxs))))
        return " ".join(t for i, t in enumerate(tokens) if i not in drop)
