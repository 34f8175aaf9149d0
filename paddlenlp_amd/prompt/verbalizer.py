"""Verbalizers: map mask-position vocabulary logits to label logits.

Reference behavior: paddlenlp/prompt/verbalizer.py (ManualVerbalizer
aggregating label-word token logits; SoftVerbalizer with a learnable head
initialized from the label words).
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.nn as nn


class ManualVerbalizer(nn.Module):
    """label_words: {label: [word, ...]}; label logit = mean over the label's
    words of the mean over each word's tokens."""

    def __init__(self, tokenizer, label_words: Dict[str, List[str]]):
        super().__init__()
        self.labels = sorted(label_words)
        token_ids = []
        for label in self.labels:
            words = label_words[label]
            word_token_ids = [tokenizer._tokenizer.encode(w).ids for w in words]
            token_ids.append(word_token_ids)
        self.token_ids = token_ids  # [n_labels][n_words][n_tokens]

    @property
    def num_labels(self):
        return len(self.labels)

    def process_logits(self, mask_logits: torch.Tensor) -> torch.Tensor:
        """mask_logits [B, V] -> label logits [B, n_labels]."""
        out = []
        for word_token_ids in self.token_ids:
            word_scores = []
            for tokens in word_token_ids:
                idx = torch.tensor(tokens, device=mask_logits.device)
                word_scores.append(mask_logits[:, idx].mean(-1))
            out.append(torch.stack(word_scores, dim=-1).mean(-1))
        return torch.stack(out, dim=-1)


class SoftVerbalizer(nn.Module):
    """Learnable label head initialized from the (first-token) label-word
    embedding rows of the MLM decoder."""

    def __init__(self, tokenizer, label_words: Dict[str, List[str]],
                 head_weight: torch.Tensor):
        super().__init__()
        self.labels = sorted(label_words)
        rows = []
        for label in self.labels:
            first_tokens = [tokenizer._tokenizer.encode(w).ids[0]
                            for w in label_words[label]]
            rows.append(head_weight[first_tokens].mean(0))
        self.head = nn.Linear(head_weight.shape[1], len(self.labels), bias=False)
        with torch.no_grad():
            self.head.weight.copy_(torch.stack(rows))

    @property
    def num_labels(self):
        return len(self.labels)

    def process_hidden(self, mask_hidden: torch.Tensor) -> torch.Tensor:
        """mask_hidden [B, H] -> label logits [B, n_labels]."""
        return self.head(mask_hidden)
