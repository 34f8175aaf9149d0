"""Tensor-parallel weight split/merge actions.

Reference: paddlenlp/transformers/conversion_utils.py:1134 ConversionMixin
(convert_tensor_parallel :1217 / merge_tensor_parallel :1251,
split_or_merge_func).  Split functions take the FULL weight and return this
rank's shard; merge functions take the list of shards and return the full
weight.  Linear weights are [out, in] (torch convention), so "column
parallel" (output split) slices dim 0 and "row parallel" (input split)
slices dim 1.
"""
from __future__ import annotations

from typing import List

import torch


def split_column(w: torch.Tensor, tp: int, rank: int) -> torch.Tensor:
    return w.chunk(tp, dim=0)[rank].contiguous()


def merge_column(shards: List[torch.Tensor]) -> torch.Tensor:
    return torch.cat(shards, dim=0)


def split_row(w: torch.Tensor, tp: int, rank: int) -> torch.Tensor:
    return w.chunk(tp, dim=1)[rank].contiguous()


def merge_row(shards: List[torch.Tensor]) -> torch.Tensor:
    return torch.cat(shards, dim=1)


def split_fused_column(w: torch.Tensor, tp: int, rank: int, parts: int = 2) -> torch.Tensor:
    """Fused [gate; up] (or qkv with equal widths): split each part, then
    re-concatenate this rank's slices so the local layout stays [gate|up]."""
    chunks = w.chunk(parts, dim=0)
    return torch.cat([c.chunk(tp, dim=0)[rank] for c in chunks], dim=0).contiguous()


def merge_fused_column(shards: List[torch.Tensor], parts: int = 2) -> torch.Tensor:
    tp = len(shards)
    per_part = [s.chunk(parts, dim=0) for s in shards]
    return torch.cat(
        [torch.cat([per_part[r][p] for r in range(tp)], dim=0) for p in range(parts)],
        dim=0,
    )
