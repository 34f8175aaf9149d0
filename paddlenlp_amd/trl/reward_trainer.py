"""Reward-model trainer (reference: llm/alignment/rm/reward_trainer.py).

Bradley-Terry pairwise loss over (chosen, rejected) scored by a causal LM
with a scalar value head on the last token.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..trainer.trainer import Trainer


class RewardModel(nn.Module):
    """Backbone + scalar head on the final non-pad position."""

    def __init__(self, backbone, hidden_size: int):
        super().__init__()
        self.backbone = backbone
        self.value_head = nn.Linear(hidden_size, 1, bias=False)

    @property
    def config(self):
        return self.backbone.config

    def score(self, input_ids, seq_lens=None):
        base = getattr(self.backbone, self.backbone.base_model_prefix, self.backbone)
        hidden = base(input_ids=input_ids)
        if isinstance(hidden, tuple):
            hidden = hidden[0]
        if seq_lens is None:
            idx = torch.full((input_ids.shape[0],), input_ids.shape[1] - 1,
                             device=input_ids.device)
        else:
            idx = (seq_lens - 1).clamp(min=0)
        last = hidden[torch.arange(hidden.shape[0], device=hidden.device), idx]
        return self.value_head(last).squeeze(-1)

    def forward(self, chosen_input_ids=None, rejected_input_ids=None,
                chosen_lens=None, rejected_lens=None, **kwargs):
        r_chosen = self.score(chosen_input_ids, chosen_lens)
        r_rejected = self.score(rejected_input_ids, rejected_lens)
        loss = -F.logsigmoid(r_chosen - r_rejected).mean()
        return loss, (r_chosen, r_rejected)


class RewardTrainer(Trainer):
    def compute_loss(self, model, inputs, return_outputs=False):
        loss, rewards = model(**inputs)
        if return_outputs:
            return loss, rewards
        return loss
