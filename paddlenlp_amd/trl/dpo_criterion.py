"""DPO loss family.

Reference behavior: paddlenlp/trl/dpo_criterion.py (296 LoC) — loss_type
dispatch at :59-97: sigmoid (w/ label smoothing), hinge, simpo, ipo, dpop,
kto_pair, sppo_hard, orpo.
"""
from __future__ import annotations

from typing import Optional, Tuple

import torch
import torch.nn as nn
import torch.nn.functional as F


class DPOCriterion(nn.Module):
    def __init__(self, beta: float = 0.1, label_smoothing: float = 0.0,
                 loss_type: str = "sigmoid", simpo_gamma: float = 0.5,
                 dpop_lambda: float = 50.0):
        super().__init__()
        self.beta = beta
        self.label_smoothing = label_smoothing
        self.loss_type = loss_type
        self.simpo_gamma = simpo_gamma
        self.dpop_lambda = dpop_lambda

    def forward(
        self,
        policy_chosen_logps: torch.Tensor,
        policy_rejected_logps: torch.Tensor,
        reference_chosen_logps: Optional[torch.Tensor] = None,
        reference_rejected_logps: Optional[torch.Tensor] = None,
    ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
        """Returns (loss, chosen_rewards, rejected_rewards)."""
        if reference_chosen_logps is None:
            reference_chosen_logps = torch.zeros_like(policy_chosen_logps)
        if reference_rejected_logps is None:
            reference_rejected_logps = torch.zeros_like(policy_rejected_logps)

        chosen_ratio = policy_chosen_logps - reference_chosen_logps
        rejected_ratio = policy_rejected_logps - reference_rejected_logps
        logits = chosen_ratio - rejected_ratio
        b = self.beta

        lt = self.loss_type
        if lt == "sigmoid":
            loss = (
                -F.logsigmoid(b * logits) * (1 - self.label_smoothing)
                - F.logsigmoid(-b * logits) * self.label_smoothing
            )
        elif lt == "hinge":
            loss = torch.relu(1 - b * logits)
        elif lt == "ipo":
            loss = (logits - 1 / (2 * b)) ** 2
        elif lt == "simpo":
            # reference-free; gamma margin (logps should be length-normalized)
            simpo_logits = policy_chosen_logps - policy_rejected_logps - self.simpo_gamma / b
            loss = -F.logsigmoid(b * simpo_logits)
        elif lt == "dpop":
            penalty = torch.clamp(reference_chosen_logps - policy_chosen_logps, min=0)
            loss = -F.logsigmoid(b * logits) + self.dpop_lambda * penalty
        elif lt == "kto_pair":
            chosen_kl = chosen_ratio.mean().clamp(min=0)
            rejected_kl = rejected_ratio.mean().clamp(min=0)
            loss = torch.cat([
                1 - torch.sigmoid(b * (chosen_ratio - rejected_kl)),
                1 - torch.sigmoid(b * (chosen_kl - rejected_ratio)),
            ])
        elif lt == "sppo_hard":
            loss = (chosen_ratio - 0.5 / b) ** 2 + (rejected_ratio + 0.5 / b) ** 2
        elif lt == "orpo":
            # reference-free odds-ratio penalty added to the NLL elsewhere
            log_odds = (policy_chosen_logps - policy_rejected_logps) - (
                torch.log1p(-torch.exp(policy_chosen_logps.clamp(max=-1e-6)))
                - torch.log1p(-torch.exp(policy_rejected_logps.clamp(max=-1e-6)))
            )
            loss = -F.logsigmoid(log_odds)
        else:
            raise ValueError(f"Unknown dpo loss_type {lt}")

        chosen_rewards = b * chosen_ratio.detach()
        rejected_rewards = b * rejected_ratio.detach()
        return loss.mean(), chosen_rewards, rejected_rewards
