"""ZeroPadding: pack SFT samples to max_length with FlashMask row indices.

Reference behavior: paddlenlp/datasets/zero_padding_dataset.py:41,106,176 —
samples are greedily packed into max_length buffers; attention isolation
between packed samples is expressed as `attn_mask_startend_row_indices`
(the FlashMask sparse form: key j may attend to queries j <= i < start[j]).
"""
from __future__ import annotations

from typing import List

import numpy as np
import torch


def generate_startend_row_indices(seq_boundaries: List[int], max_length: int) -> np.ndarray:
    """[1, S, 1] int32: for key position j inside sample (b0, b1), queries in
    [j, b1) can see it — causal within the sample, blind across samples."""
    idx = np.zeros((1, max_length, 1), dtype=np.int32)
    prev = 0
    for b in seq_boundaries:
        idx[0, prev:b, 0] = b
        prev = b
    idx[0, prev:, 0] = prev  # padding region sees nothing
    return idx


class ZeroPaddingMapDataset(torch.utils.data.Dataset):
    """Pack a map-style dataset of tokenized samples.

    Each input sample: {"input_ids": [...], "labels": [...]}.
    Output: packed {"input_ids", "labels", "attn_mask_startend_row_indices"}.
    """

    def __init__(self, data, tokenizer=None, max_length: int = 2048, greedy: bool = False):
        self.max_length = max_length
        samples = list(data)
        if greedy:
            samples = sorted(samples, key=lambda ex: -len(ex["input_ids"]))
        self.packs = self._pack(samples)

    def _pack(self, samples):
        packs = []
        cur_ids: List[int] = []
        cur_labels: List[int] = []
        boundaries: List[int] = []
        for ex in samples:
            ids = list(ex["input_ids"])[: self.max_length]
            labels = list(ex.get("labels", ids))[: self.max_length]
            if len(cur_ids) + len(ids) > self.max_length and cur_ids:
                packs.append(self._finish(cur_ids, cur_labels, boundaries))
                cur_ids, cur_labels, boundaries = [], [], []
            cur_ids.extend(ids)
            cur_labels.extend(labels)
            boundaries.append(len(cur_ids))
        if cur_ids:
            packs.append(self._finish(cur_ids, cur_labels, boundaries))
        return packs

    def _finish(self, ids, labels, boundaries):
        pad = self.max_length - len(ids)
        return {
            "input_ids": np.asarray(ids + [0] * pad, dtype=np.int64),
            "labels": np.asarray(labels + [-100] * pad, dtype=np.int64),
            "attn_mask_startend_row_indices": generate_startend_row_indices(boundaries, self.max_length),
        }

    def __len__(self):
        return len(self.packs)

    def __getitem__(self, idx):
        return self.packs[idx]


ZeroPaddingIterableDataset = ZeroPaddingMapDataset  # map version covers both uses here
