"""Dataclass model outputs (reference: paddlenlp/transformers/model_outputs.py)."""
from __future__ import annotations

from dataclasses import dataclass, fields
from typing import Any, Optional, Tuple

import torch


@dataclass
class ModelOutput:
    """Tuple-compatible dataclass base (index and attribute access)."""

    def __getitem__(self, idx):
        vals = [getattr(self, f.name) for f in fields(self) if getattr(self, f.name) is not None]
        return vals[idx]

    def __iter__(self):
        return iter([getattr(self, f.name) for f in fields(self)
                     if getattr(self, f.name) is not None])

    def to_tuple(self) -> Tuple:
        return tuple(getattr(self, f.name) for f in fields(self))


@dataclass
class BaseModelOutputWithPast(ModelOutput):
    last_hidden_state: Optional[torch.Tensor] = None
    past_key_values: Optional[Any] = None
    hidden_states: Optional[Tuple[torch.Tensor, ...]] = None
    attentions: Optional[Tuple[torch.Tensor, ...]] = None


@dataclass
class CausalLMOutputWithPast(ModelOutput):
    loss: Optional[torch.Tensor] = None
    logits: Optional[torch.Tensor] = None
    past_key_values: Optional[Any] = None
    hidden_states: Optional[Tuple[torch.Tensor, ...]] = None
    attentions: Optional[Tuple[torch.Tensor, ...]] = None


@dataclass
class CausalLMOutputWithCrossAttentions(CausalLMOutputWithPast):
    cross_attentions: Optional[Tuple[torch.Tensor, ...]] = None


@dataclass
class SequenceClassifierOutput(ModelOutput):
    loss: Optional[torch.Tensor] = None
    logits: Optional[torch.Tensor] = None
    hidden_states: Optional[Tuple[torch.Tensor, ...]] = None
