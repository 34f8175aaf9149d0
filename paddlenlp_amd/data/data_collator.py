"""Data collators (reference: paddlenlp/data/data_collator.py)."""
from __future__ import annotations

from dataclasses import dataclass
from typing import Any, Dict, List, Optional

import numpy as np
import torch


def default_data_collator(features: List[Dict[str, Any]]) -> Dict[str, torch.Tensor]:
    if not features:
        return {}
    batch = {}
    first = features[0]
    for key, value in first.items():
        if value is None:
            continue
        vals = [f[key] for f in features]
        if isinstance(value, torch.Tensor):
            batch[key] = torch.stack(vals)
        elif isinstance(value, np.ndarray):
            batch[key] = torch.from_numpy(np.stack(vals))
        elif isinstance(value, (int, float, list)):
            batch[key] = torch.tensor(vals)
        else:
            batch[key] = vals
    return batch


@dataclass
class DataCollatorWithPadding:
    tokenizer: Any = None
    pad_to_multiple_of: Optional[int] = None
    max_length: Optional[int] = None
    label_pad_token_id: int = -100

    def _pad_len(self, longest: int) -> int:
        length = self.max_length or longest
        if self.pad_to_multiple_of:
            length = ((length + self.pad_to_multiple_of - 1) // self.pad_to_multiple_of) * self.pad_to_multiple_of
        return length

    def __call__(self, features: List[Dict[str, Any]]) -> Dict[str, torch.Tensor]:
        pad_id = 0
        if self.tokenizer is not None and self.tokenizer.pad_token_id is not None:
            pad_id = self.tokenizer.pad_token_id
        keys = features[0].keys()
        longest = max(len(f["input_ids"]) for f in features)
        length = self._pad_len(longest)
        batch = {}
        for key in keys:
            fill = self.label_pad_token_id if key == "labels" else (pad_id if key == "input_ids" else 0)
            rows = []
            for f in features:
                row = list(f[key])
                row = row + [fill] * (length - len(row))
                rows.append(row[:length])
            batch[key] = torch.tensor(rows)
        if "attention_mask" not in batch:
            batch["attention_mask"] = (batch["input_ids"] != pad_id).long()
        return batch


@dataclass
class DataCollatorForSeq2Seq(DataCollatorWithPadding):
    pass


@dataclass
class DataCollatorForLanguageModeling:
    """Causal-LM collator: labels = input_ids shifted inside the model/loss.

    We follow the reference pretrain convention (GPTDataset returns
    input_ids[0:s] + labels[1:s+1] already shifted)."""

    tokenizer: Any = None
    return_tensors: str = "pt"

    def __call__(self, features):
        return default_data_collator(features)
