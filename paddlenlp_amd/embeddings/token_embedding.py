"""Static token embeddings with a vocabulary and OOV handling.

Reference behavior: paddlenlp/embeddings/token_embedding.py:40
(TokenEmbedding over pretrained word-vector files with search/cosine-sim
helpers).  There is no network in this environment, so embeddings load from
a local .npz/.txt table or random-init from a vocab list.
"""
from __future__ import annotations

import os
from typing import List, Optional

import numpy as np
import torch
import torch.nn as nn
import torch.nn.functional as F

PAD_TOKEN = "[PAD]"
UNK_TOKEN = "[UNK]"


class TokenEmbedding(nn.Embedding):
    def __init__(self, embedding_source: Optional[str] = None,
                 vocab: Optional[List[str]] = None, embedding_dim: int = 300,
                 trainable: bool = True, keep_extended_vocab_only: bool = False):
        if embedding_source is not None:
            words, table = self._load_table(embedding_source)
        else:
            assert vocab is not None, "need embedding_source or vocab"
            words = list(vocab)
            rng = np.random.RandomState(0)
            table = rng.normal(scale=0.02,
                               size=(len(words), embedding_dim)).astype(np.float32)
        # dedupe keeping the first occurrence, rows aligned to final indices
        self._word_to_idx = {}
        rows = []
        for w, row in zip(words, table):
            if w not in self._word_to_idx:
                self._word_to_idx[w] = len(self._word_to_idx)
                rows.append(row)
        for special in (PAD_TOKEN, UNK_TOKEN):
            if special not in self._word_to_idx:
                self._word_to_idx[special] = len(self._word_to_idx)
                rows.append(np.zeros(table.shape[1], np.float32))
        vectors = np.stack(rows)
        self._idx_to_word = {i: w for w, i in self._word_to_idx.items()}
        super().__init__(len(self._word_to_idx), vectors.shape[1],
                         padding_idx=self._word_to_idx[PAD_TOKEN])
        with torch.no_grad():
            self.weight.copy_(torch.from_numpy(vectors))
        self.weight.requires_grad_(trainable)
        self.unk_idx = self._word_to_idx[UNK_TOKEN]

    @staticmethod
    def _load_table(path: str):
        if path.endswith(".npz"):
            data = np.load(path, allow_pickle=True)
            return list(data["vocab"]), data["embedding"].astype(np.float32)
        words, vecs = [], []
        with open(path) as f:
            for line in f:
                parts = line.rstrip("\n").split()
                if len(parts) < 3:
                    continue  # header / malformed
                words.append(parts[0])
                vecs.append([float(x) for x in parts[1:]])
        return words, np.asarray(vecs, dtype=np.float32)

    # ------------------------------------------------------------- lookups
    def get_idx_from_word(self, word: str) -> int:
        return self._word_to_idx.get(word, self.unk_idx)

    def get_idx_list_from_words(self, words: List[str]) -> List[int]:
        return [self.get_idx_from_word(w) for w in words]

    def search(self, words) -> np.ndarray:
        if isinstance(words, str):
            words = [words]
        idx = torch.tensor(self.get_idx_list_from_words(words))
        return self.weight.detach()[idx].numpy()

    def cosine_sim(self, word_a: str, word_b: str) -> float:
        va, vb = self.search(word_a)[0], self.search(word_b)[0]
        return float(F.cosine_similarity(torch.from_numpy(va)[None],
                                         torch.from_numpy(vb)[None]))

    def dot(self, word_a: str, word_b: str) -> float:
        va, vb = self.search(word_a)[0], self.search(word_b)[0]
        return float(np.dot(va, vb))

    @property
    def vocab_size(self):
        return self.num_embeddings
