"""Weighted multi-corpus mixing (reference: paddlenlp/data/blendable_dataset.py)."""
from __future__ import annotations

from typing import List

import numpy as np
import torch


class BlendableDataset(torch.utils.data.Dataset):
    def __init__(self, datasets: List, weights: List[float], size: int):
        self.datasets = [d for d in datasets if d is not None]
        weights = [w for d, w in zip(datasets, weights) if d is not None]
        s = sum(weights)
        self.weights = [w / s for w in weights]
        self.size = size

        # per-sample (dataset_index, sample_index) maps, weight-proportional
        n = len(self.datasets)
        self.dataset_index = np.zeros(size, dtype=np.int64)
        self.dataset_sample_index = np.zeros(size, dtype=np.int64)
        counts = [0] * n
        errors = [0.0] * n
        for i in range(size):
            # largest-remainder assignment keeps the realized mix on-weight
            best, best_err = 0, -1e9
            for d in range(n):
                err = self.weights[d] * (i + 1) - counts[d]
                if err > best_err:
                    best, best_err = d, err
            self.dataset_index[i] = best
            self.dataset_sample_index[i] = counts[best] % len(self.datasets[best])
            counts[best] += 1

    def __len__(self):
        return self.size

    def __getitem__(self, idx):
        d = self.dataset_index[idx]
        s = self.dataset_sample_index[idx]
        return self.datasets[d][s]
