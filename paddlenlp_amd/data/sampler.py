"""Distributed batch sampler with consumed_samples resume.

Reference: paddlenlp/data/sampler.py + paddlenlp/utils/batch_sampler.py
(NlpDistributedBatchSampler, resume via consumed_samples trainer.py:916-923).
"""
from __future__ import annotations

import math
from typing import Iterator, List

import numpy as np
import torch


class DistributedBatchSampler(torch.utils.data.Sampler):
    def __init__(
        self,
        dataset,
        batch_size: int,
        num_replicas: int = 1,
        rank: int = 0,
        shuffle: bool = False,
        drop_last: bool = True,
        seed: int = 42,
        consumed_samples: int = 0,
    ):
        self.dataset = dataset
        self.batch_size = batch_size
        self.num_replicas = num_replicas
        self.rank = rank
        self.shuffle = shuffle
        self.drop_last = drop_last
        self.seed = seed
        self.epoch = 0
        self.consumed_samples = consumed_samples

        total = len(dataset)
        if drop_last:
            self.num_samples = total // num_replicas
        else:
            self.num_samples = math.ceil(total / num_replicas)
        self.total_size = self.num_samples * num_replicas

    def set_epoch(self, epoch: int):
        self.epoch = epoch

    def __iter__(self) -> Iterator[List[int]]:
        if self.shuffle:
            g = np.random.default_rng(self.seed + self.epoch)
            indices = g.permutation(len(self.dataset)).tolist()
        else:
            indices = list(range(len(self.dataset)))
        if not self.drop_last:
            pad = self.total_size - len(indices)
            indices += indices[:pad]
        indices = indices[: self.total_size]

        # skip already-consumed samples (resume)
        skip = (self.consumed_samples // self.num_replicas) % max(1, self.num_samples)
        local = indices[self.rank:self.total_size:self.num_replicas]
        local = local[skip:]

        batch = []
        for idx in local:
            batch.append(idx)
            if len(batch) == self.batch_size:
                yield batch
                batch = []
        if batch and not self.drop_last:
            yield batch

    def __len__(self):
        n = self.num_samples - (self.consumed_samples // self.num_replicas) % max(1, self.num_samples)
        if self.drop_last:
            return n // self.batch_size
        return math.ceil(n / self.batch_size)
