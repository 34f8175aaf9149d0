"""Classic batchify helpers: Stack / Pad / Tuple / Dict.

Reference behavior: paddlenlp/data/collate.py:26-300 (the composable
collate functions the classic example pipelines use).
"""
from __future__ import annotations

from typing import Callable, List, Optional

import numpy as np


class Stack:
    """Stack equal-shape samples into one array."""

    def __init__(self, axis: int = 0, dtype=None):
        self.axis = axis
        self.dtype = dtype

    def __call__(self, data: List):
        arr = np.stack([np.asarray(d) for d in data], axis=self.axis)
        return arr.astype(self.dtype) if self.dtype else arr


class Pad:
    """Pad variable-length samples to the batch max (or a fixed axis size)."""

    def __init__(self, pad_val=0, axis: int = 0, ret_length: bool = False,
                 dtype=None, pad_right: bool = True):
        self.pad_val = pad_val
        self.axis = axis
        self.ret_length = ret_length
        self.dtype = dtype
        self.pad_right = pad_right

    def __call__(self, data: List):
        arrs = [np.asarray(d) for d in data]
        max_len = max(a.shape[self.axis] for a in arrs)
        out = []
        lengths = []
        for a in arrs:
            lengths.append(a.shape[self.axis])
            pad_width = [(0, 0)] * a.ndim
            pad = max_len - a.shape[self.axis]
            pad_width[self.axis] = (0, pad) if self.pad_right else (pad, 0)
            out.append(np.pad(a, pad_width, constant_values=self.pad_val))
        batch = np.stack(out)
        if self.dtype:
            batch = batch.astype(self.dtype)
        if self.ret_length:
            return batch, np.asarray(lengths, dtype="int64")
        return batch


class Tuple:
    """Apply the i-th function to the i-th field of each sample."""

    def __init__(self, fn, *args):
        if isinstance(fn, (list, tuple)):
            assert not args, "pass a single list or varargs, not both"
            self._fn = list(fn)
        else:
            self._fn = [fn] + list(args)

    def __call__(self, data: List):
        assert len(data[0]) == len(self._fn), (
            f"{len(self._fn)} collate fns for {len(data[0])} fields")
        out = []
        for i, fn in enumerate(self._fn):
            result = fn([sample[i] for sample in data])
            if isinstance(result, (tuple, list)):
                out.extend(result)
            else:
                out.append(result)
        return tuple(out)


class Dict:
    """Apply per-key functions to dict samples: {key: fn}."""

    def __init__(self, fn: dict):
        self._fn = fn

    def __call__(self, data: List[dict]):
        out = []
        for key, fn in self._fn.items():
            result = fn([sample[key] for sample in data])
            if isinstance(result, (tuple, list)):
                out.extend(result)
            else:
                out.append(result)
        return tuple(out)


class Vocab:
    """Token <-> id mapping with specials (reference data/vocab.py:24)."""

    def __init__(self, counter=None, max_size: Optional[int] = None,
                 min_freq: int = 1, token_to_idx: Optional[dict] = None,
                 unk_token: Optional[str] = None,
                 pad_token: Optional[str] = None,
                 bos_token: Optional[str] = None,
                 eos_token: Optional[str] = None):
        self.unk_token = unk_token
        self.pad_token = pad_token
        self.bos_token = bos_token
        self.eos_token = eos_token
        specials = [t for t in (pad_token, unk_token, bos_token, eos_token)
                    if t is not None]
        if token_to_idx is not None:
            self._token_to_idx = dict(token_to_idx)
        else:
            self._token_to_idx = {}
            for t in specials:
                if t not in self._token_to_idx:
                    self._token_to_idx[t] = len(self._token_to_idx)
            if counter:
                items = sorted(counter.items(), key=lambda kv: (-kv[1], kv[0]))
                for tok, freq in items:
                    if freq < min_freq or tok in self._token_to_idx:
                        continue
                    if max_size and len(self._token_to_idx) >= max_size:
                        break
                    self._token_to_idx[tok] = len(self._token_to_idx)
        self._idx_to_token = {i: t for t, i in self._token_to_idx.items()}

    def __len__(self):
        return len(self._token_to_idx)

    def __contains__(self, token):
        return token in self._token_to_idx

    @property
    def token_to_idx(self):
        return self._token_to_idx

    def to_indices(self, tokens):
        unk = self._token_to_idx.get(self.unk_token)
        if isinstance(tokens, str):
            return self._token_to_idx.get(tokens, unk)
        return [self._token_to_idx.get(t, unk) for t in tokens]

    def to_tokens(self, indices):
        if isinstance(indices, int):
            return self._idx_to_token[indices]
        return [self._idx_to_token[int(i)] for i in indices]

    @classmethod
    def build_vocab(cls, iterator, **kwargs):
        from collections import Counter

        counter = Counter()
        for tokens in iterator:
            counter.update(tokens)
        return cls(counter=counter, **kwargs)


class JiebaLikeTokenizer:
    """Greedy longest-match word tokenizer over a vocab (reference
    data/tokenizer.py JiebaTokenizer's role, without the jieba dependency —
    no external dictionaries offline)."""

    def __init__(self, vocab: Vocab, max_word_len: int = 8):
        self.vocab = vocab
        self.max_word_len = max_word_len

    def cut(self, sentence: str) -> List[str]:
        out = []
        i = 0
        n = len(sentence)
        while i < n:
            match = None
            for L in range(min(self.max_word_len, n - i), 0, -1):
                piece = sentence[i:i + L]
                if piece in self.vocab:
                    match = piece
                    break
            if match is None:
                match = sentence[i]
            out.append(match)
            i += len(match)
        return out

    def encode(self, sentence: str) -> List[int]:
        return self.vocab.to_indices(self.cut(sentence))
