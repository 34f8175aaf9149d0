"""Megatron-style GPT pretraining dataset over MMapIndexedDataset.

Reference behavior: paddlenlp/data/causal_dataset.py — split string
"949,50,1" -> get_train_valid_test_split_ :49, build_train_valid_test_datasets
:112, GPTDataset :282 with doc/sample/shuffle index npy mmaps :336-342,
_build_index_mappings :417 (rank 0 builds, others wait on the filesystem).
"""
from __future__ import annotations

import hashlib
import os
import time
from typing import List, Optional

import numpy as np
import torch

from ..utils.log import logger
from .indexed_dataset import make_indexed_dataset


def get_train_valid_test_split_(splits_string: str, size: int) -> List[int]:
    """'949,50,1' -> cumulative [0, a, b, size] boundaries."""
    splits = [float(s) for s in splits_string.replace("/", ",").split(",")]
    while len(splits) < 3:
        splits.append(0.0)
    splits = splits[:3]
    total = sum(splits)
    assert total > 0
    weights = [s / total for s in splits]
    idx = [0]
    for w in weights:
        idx.append(idx[-1] + int(round(w * size)))
    diff = idx[-1] - size
    idx[-1] -= diff
    return idx


def build_train_valid_test_datasets(
    data_prefix,
    splits_string: str,
    train_val_test_num_samples,
    seq_length: int,
    seed: int,
    data_cache_path: Optional[str] = None,
):
    """Single- or multi-corpus (weighted) dataset triple."""
    if isinstance(data_prefix, (list, tuple)) and len(data_prefix) > 1:
        # "weight1 prefix1 weight2 prefix2" style list
        if len(data_prefix) % 2 == 0 and _looks_weighted(data_prefix):
            from .blendable_dataset import BlendableDataset

            weights = [float(data_prefix[i]) for i in range(0, len(data_prefix), 2)]
            prefixes = [data_prefix[i + 1] for i in range(0, len(data_prefix), 2)]
            s = sum(weights)
            weights = [w / s for w in weights]
            trains, valids, tests = [], [], []
            for prefix in prefixes:
                t, v, te = _build_single(prefix, splits_string,
                                         train_val_test_num_samples, seq_length,
                                         seed, data_cache_path)
                trains.append(t); valids.append(v); tests.append(te)
            train = BlendableDataset(trains, weights, train_val_test_num_samples[0])
            valid = BlendableDataset(valids, weights, train_val_test_num_samples[1])
            test = BlendableDataset(tests, weights, train_val_test_num_samples[2])
            return train, valid, test
        data_prefix = data_prefix[0]
    if isinstance(data_prefix, (list, tuple)):
        data_prefix = data_prefix[0]
    return _build_single(data_prefix, splits_string, train_val_test_num_samples,
                         seq_length, seed, data_cache_path)


def _looks_weighted(parts):
    try:
        float(parts[0])
        return True
    except (TypeError, ValueError):
        return False


def _build_single(data_prefix, splits_string, num_samples, seq_length, seed, cache):
    indexed = make_indexed_dataset(data_prefix)
    total_docs = len(indexed.doc_idx) - 1
    splits = get_train_valid_test_split_(splits_string, total_docs)
    names = ["train", "valid", "test"]
    out = []
    for i, name in enumerate(names):
        if splits[i + 1] > splits[i] and (num_samples[i] or 0) > 0:
            documents = np.arange(splits[i], splits[i + 1], dtype=np.int32)
            out.append(GPTDataset(name, data_prefix, documents, indexed,
                                  num_samples[i], seq_length, seed, cache))
        else:
            out.append(None)
    return tuple(out)


class GPTDataset(torch.utils.data.Dataset):
    """Token stream chunked into seq_length+1 samples via the Megatron
    doc/sample/shuffle index triple (reference GPTDataset :282)."""

    def __init__(self, name, data_prefix, documents, indexed_dataset,
                 num_samples, seq_length, seed, data_cache_path=None):
        self.name = name
        self.indexed_dataset = indexed_dataset
        self.seq_length = seq_length
        self.doc_idx, self.sample_idx, self.shuffle_idx = _build_index_mappings(
            name, data_prefix, documents, indexed_dataset.sizes,
            num_samples, seq_length, seed, data_cache_path,
        )

    def __len__(self):
        return self.sample_idx.shape[0] - 1

    def __getitem__(self, idx):
        idx = self.shuffle_idx[idx]
        doc_index_f = self.sample_idx[idx][0]
        doc_index_l = self.sample_idx[idx + 1][0]
        offset_f = self.sample_idx[idx][1]
        offset_l = self.sample_idx[idx + 1][1]
        if doc_index_f == doc_index_l:
            sample = self.indexed_dataset.get(
                self.doc_idx[doc_index_f], offset=offset_f,
                length=offset_l - offset_f + 1)
        else:
            parts = [self.indexed_dataset.get(self.doc_idx[doc_index_f], offset=offset_f)]
            for i in range(doc_index_f + 1, doc_index_l):
                parts.append(self.indexed_dataset.get(self.doc_idx[i]))
            parts.append(self.indexed_dataset.get(self.doc_idx[doc_index_l],
                                                  length=offset_l + 1))
            sample = np.concatenate(parts)
        tokens = np.asarray(sample, dtype=np.int64)
        return {
            "input_ids": tokens[:-1],
            "labels": tokens[1:],
        }


def _build_index_mappings(name, data_prefix, documents, sizes, num_samples,
                          seq_length, seed, data_cache_path):
    """doc_idx / sample_idx / shuffle_idx (cached as .npy next to the data;
    rank 0 builds, other ranks wait — reference :417)."""
    tokens_per_epoch = int(np.sum(sizes[documents]))
    num_epochs = max(1, int(np.ceil((num_samples * seq_length + 1) / tokens_per_epoch)))

    cache_dir = data_cache_path or (os.path.dirname(data_prefix) or ".")
    tag = hashlib.md5(
        f"{name}-{len(documents)}-{num_epochs}-{seq_length}-{seed}-{num_samples}".encode()
    ).hexdigest()[:16]
    base = os.path.join(cache_dir, f"{os.path.basename(data_prefix)}_{name}_{tag}")
    doc_f, samp_f, shuf_f = base + "_doc.npy", base + "_sample.npy", base + "_shuffle.npy"

    is_rank0 = (not torch.distributed.is_initialized()) or torch.distributed.get_rank() == 0
    if is_rank0 and not (os.path.isfile(doc_f) and os.path.isfile(samp_f) and os.path.isfile(shuf_f)):
        rng = np.random.RandomState(seed)
        # doc_idx: documents repeated num_epochs times, shuffled
        doc_idx = np.tile(documents, num_epochs)
        rng.shuffle(doc_idx)
        np.save(doc_f, doc_idx, allow_pickle=False)

        # sample_idx: [n_samples+1, 2] (doc position, offset in doc)
        total_tokens = tokens_per_epoch * num_epochs
        n_samples = (total_tokens - 1) // seq_length
        sample_idx = np.zeros((n_samples + 1, 2), dtype=np.int64)
        doc_pos, offset = 0, 0
        sample_idx[0] = (0, 0)
        for i in range(1, n_samples + 1):
            remaining = seq_length
            while remaining > 0:
                doc_len = sizes[doc_idx[doc_pos]] - offset
                if doc_len > remaining:
                    offset += remaining
                    remaining = 0
                else:
                    remaining -= doc_len
                    doc_pos += 1
                    offset = 0
                    if doc_pos >= len(doc_idx):
                        doc_pos = len(doc_idx) - 1
                        offset = sizes[doc_idx[doc_pos]] - 1
                        remaining = 0
            sample_idx[i] = (doc_pos, offset)
        np.save(samp_f, sample_idx, allow_pickle=False)

        shuffle_idx = np.arange(n_samples, dtype=np.int64)
        rng.shuffle(shuffle_idx)
        np.save(shuf_f, shuffle_idx, allow_pickle=False)
        logger.info(f"[{name}] built index mappings: {n_samples} samples, {num_epochs} epochs")

    if torch.distributed.is_initialized():
        torch.distributed.barrier()
    # wait for rank 0 on shared filesystems
    for _ in range(600):
        if os.path.isfile(doc_f) and os.path.isfile(samp_f) and os.path.isfile(shuf_f):
            break
        time.sleep(0.5)

    doc_idx = np.load(doc_f, mmap_mode="r")
    sample_idx = np.load(samp_f, mmap_mode="r")
    shuffle_idx = np.load(shuf_f, mmap_mode="r")
    return doc_idx, sample_idx, shuffle_idx
