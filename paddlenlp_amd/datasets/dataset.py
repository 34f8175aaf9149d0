"""load_dataset registry + MapDataset (reference: paddlenlp/datasets/dataset.py).

Local-file focused (no network): json/jsonl files with one example per line,
or a user-provided read function.
"""
from __future__ import annotations

import json
import os
from typing import Callable, Dict, List

import torch


class MapDataset(torch.utils.data.Dataset):
    def __init__(self, data: List):
        self.data = list(data)
        self._transforms = []

    def __len__(self):
        return len(self.data)

    def __getitem__(self, idx):
        ex = self.data[idx]
        for fn in self._transforms:
            ex = fn(ex)
        return ex

    def map(self, fn: Callable, lazy: bool = False):
        if lazy:
            self._transforms.append(fn)
        else:
            self.data = [fn(ex) for ex in self.data]
        return self

    def filter(self, fn: Callable):
        self.data = [ex for ex in self.data if fn(ex)]
        return self


def _read_json_file(path: str) -> List[Dict]:
    examples = []
    with open(path, "r", encoding="utf-8") as f:
        content = f.read().strip()
    if content.startswith("["):
        return json.loads(content)
    for line in content.splitlines():
        line = line.strip()
        if line:
            examples.append(json.loads(line))
    return examples


def load_dataset(path_or_read_fn, data_files=None, splits=None, lazy=False, **kwargs):
    """Local loader: a directory with {split}.json, an explicit file list,
    or a callable yielding examples."""
    if callable(path_or_read_fn):
        return MapDataset(list(path_or_read_fn(**kwargs)))

    path = path_or_read_fn
    if data_files is not None:
        if isinstance(data_files, str):
            return MapDataset(_read_json_file(data_files))
        return [MapDataset(_read_json_file(f)) for f in data_files]

    if os.path.isfile(path):
        return MapDataset(_read_json_file(path))

    if os.path.isdir(path):
        splits = splits or ["train", "dev", "test"]
        if isinstance(splits, str):
            splits = [splits]
        out = []
        for split in splits:
            for ext in (".json", ".jsonl"):
                f = os.path.join(path, split + ext)
                if os.path.isfile(f):
                    out.append(MapDataset(_read_json_file(f)))
                    break
            else:
                out.append(None)
        return out[0] if len(out) == 1 else out
    raise FileNotFoundError(f"Cannot load dataset from {path} (no network access)")
