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
class DataCollatorForTokenClassification(DataCollatorWithPadding):
    """Pads labels with -100 alongside input_ids (reference :216)."""


@dataclass
class DataCollatorForLanguageModeling:
    """LM collator (reference :501).  mlm=False: causal convention —
    labels are the (already next-token-shifted) input ids, straight
    stacking.  mlm=True: BERT masking — each non-special token is
    selected with `mlm_probability`; of those 80% -> [MASK], 10% ->
    random token, 10% unchanged; unselected positions get label -100."""

    tokenizer: Any = None
    mlm: bool = False
    mlm_probability: float = 0.15
    return_tensors: str = "pt"

    def _mask(self, ids: torch.Tensor):
        labels = ids.clone()
        prob = torch.full(ids.shape, self.mlm_probability)
        special = torch.zeros_like(ids, dtype=torch.bool)
        tok = self.tokenizer
        if tok is not None:
            for t in (tok.pad_token_id, tok.cls_token_id, tok.sep_token_id):
                if t is not None:
                    special |= ids == t
        prob.masked_fill_(special, 0.0)
        selected = torch.bernoulli(prob).bool()
        labels[~selected] = -100
        mask_id = (tok.mask_token_id if tok is not None and
                   tok.mask_token_id is not None else 0)
        vocab = (len(tok) if tok is not None else int(ids.max()) + 1)
        replace = torch.bernoulli(torch.full(ids.shape, 0.8)).bool() & selected
        ids[replace] = mask_id
        random = (torch.bernoulli(torch.full(ids.shape, 0.5)).bool()
                  & selected & ~replace)
        ids[random] = torch.randint(vocab, ids.shape)[random]
        return ids, labels

    def __call__(self, features):
        batch = default_data_collator(features)
        if self.mlm:
            ids, labels = self._mask(batch["input_ids"].clone())
            batch["input_ids"] = ids
            batch["labels"] = labels
        return batch


class DataCollatorForWholeWordMask(DataCollatorForLanguageModeling):
    """Whole-word masking (reference :656): tokens carrying a `##`
    continuation prefix are masked together with their word head."""

    def __init__(self, tokenizer=None, mlm_probability=0.15):
        super().__init__(tokenizer=tokenizer, mlm=True,
                         mlm_probability=mlm_probability)

    def _mask(self, ids: torch.Tensor):
        tok = self.tokenizer
        labels = ids.clone()
        mask_id = (tok.mask_token_id if tok is not None and
                   tok.mask_token_id is not None else 0)
        B, S = ids.shape
        selected = torch.zeros_like(ids, dtype=torch.bool)
        for b in range(B):
            # group positions into words via the ## continuation marker
            words, cur = [], []
            for i in range(S):
                piece = (tok.convert_ids_to_tokens([int(ids[b, i])])[0]
                         if tok is not None else str(int(ids[b, i])))
                if piece.startswith("##") and cur:
                    cur.append(i)
                else:
                    if cur:
                        words.append(cur)
                    cur = [i]
            if cur:
                words.append(cur)
            n_mask = max(1, int(len(words) * self.mlm_probability))
            order = torch.randperm(len(words))[:n_mask]
            for w in order:
                for i in words[int(w)]:
                    selected[b, i] = True
        labels[~selected] = -100
        ids = ids.clone()
        ids[selected] = mask_id
        return ids, labels
