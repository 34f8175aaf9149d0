"""R-Drop regularization loss (reference: paddlenlp/losses/rdrop.py)."""
from __future__ import annotations

import torch
import torch.nn.functional as F


class RDropLoss(torch.nn.Module):
    """Symmetric KL between two stochastic forward passes."""

    def __init__(self, reduction: str = "none"):
        super().__init__()
        self.reduction = reduction

    def forward(self, p_logits: torch.Tensor, q_logits: torch.Tensor,
                pad_mask: torch.Tensor = None) -> torch.Tensor:
        p = p_logits.float().log_softmax(-1)
        q = q_logits.float().log_softmax(-1)
        kl_pq = F.kl_div(q, p.exp(), reduction="none").sum(-1)
        kl_qp = F.kl_div(p, q.exp(), reduction="none").sum(-1)
        loss = 0.5 * (kl_pq + kl_qp)
        if pad_mask is not None:
            loss = loss * pad_mask
            return loss.sum() / pad_mask.sum().clamp(min=1)
        if self.reduction == "mean":
            return loss.mean()
        if self.reduction == "sum":
            return loss.sum()
        return loss
