"""Linear-chain CRF layer + Viterbi decoder.

Reference behavior: paddlenlp/layers/crf.py (LinearChainCrf,
LinearChainCrfLoss, ViterbiDecoder).  Transitions are stored with two extra
virtual tags appended: [num_labels] = START, [num_labels+1] = STOP, matching
the reference's ``with_start_stop_tag`` layout.  Everything is batched
tensor code (forward-algorithm log-partition, gold-path score, Viterbi
max-product) — no per-token Python loops over the batch.
"""
from __future__ import annotations

import torch
import torch.nn as nn

__all__ = ["LinearChainCrf", "LinearChainCrfLoss", "ViterbiDecoder"]


class LinearChainCrf(nn.Module):
    def __init__(self, num_labels: int, crf_lr: float = 0.1,
                 with_start_stop_tag: bool = True):
        super().__init__()
        self.num_labels = num_labels
        self.with_start_stop_tag = with_start_stop_tag
        n = num_labels + 2 if with_start_stop_tag else num_labels
        self.num_tags = n
        self.transitions = nn.Parameter(crf_lr * torch.randn(n, n))
        if with_start_stop_tag:
            self.start_idx = num_labels
            self.stop_idx = num_labels + 1

    def _lengths_to_mask(self, lengths, max_len):
        ar = torch.arange(max_len, device=lengths.device)[None, :]
        return ar < lengths[:, None]

    def forward(self, inputs: torch.Tensor, lengths: torch.Tensor):
        """Log-partition log Z via the forward algorithm.

        inputs: [B, S, num_labels] emission scores; lengths: [B]."""
        B, S, L = inputs.shape
        mask = self._lengths_to_mask(lengths, S)
        trans = self.transitions[:L, :L]  # tag->tag
        if self.with_start_stop_tag:
            alpha = inputs[:, 0] + self.transitions[self.start_idx, :L][None, :]
        else:
            alpha = inputs[:, 0]
        for t in range(1, S):
            # [B, from, to]: alpha + transition + emission(to)
            scores = alpha[:, :, None] + trans[None, :, :] + inputs[:, t, None, :]
            new_alpha = torch.logsumexp(scores, dim=1)
            alpha = torch.where(mask[:, t, None], new_alpha, alpha)
        if self.with_start_stop_tag:
            alpha = alpha + self.transitions[:L, self.stop_idx][None, :]
        return torch.logsumexp(alpha, dim=1)

    def gold_score(self, inputs: torch.Tensor, labels: torch.Tensor,
                   lengths: torch.Tensor):
        """Score of the gold path: sum emissions + sum transitions."""
        B, S, L = inputs.shape
        mask = self._lengths_to_mask(lengths, S).to(inputs.dtype)
        emis = inputs.gather(2, labels.unsqueeze(-1)).squeeze(-1)  # [B,S]
        score = (emis * mask).sum(1)
        prev, nxt = labels[:, :-1], labels[:, 1:]
        trans_scores = self.transitions[prev, nxt]  # [B, S-1]
        score = score + (trans_scores * mask[:, 1:]).sum(1)
        if self.with_start_stop_tag:
            score = score + self.transitions[self.start_idx, labels[:, 0]]
            last = labels.gather(1, (lengths - 1).clamp(min=0)[:, None]).squeeze(1)
            score = score + self.transitions[last, self.stop_idx]
        return score


class LinearChainCrfLoss(nn.Module):
    """NLL = logZ - gold_score, averaged over the batch."""

    def __init__(self, crf: LinearChainCrf):
        super().__init__()
        self.crf = crf

    def forward(self, inputs, lengths, labels):
        return (self.crf(inputs, lengths)
                - self.crf.gold_score(inputs, labels, lengths)).mean()


class ViterbiDecoder(nn.Module):
    def __init__(self, transitions: torch.Tensor,
                 with_start_stop_tag: bool = True):
        super().__init__()
        self.transitions = transitions
        self.with_start_stop_tag = with_start_stop_tag

    def forward(self, inputs: torch.Tensor, lengths: torch.Tensor):
        """Returns (scores [B], paths [B, S])."""
        B, S, L = inputs.shape
        trans = self.transitions[:L, :L]
        if self.with_start_stop_tag:
            start_idx, stop_idx = L, L + 1
            alpha = inputs[:, 0] + self.transitions[start_idx, :L][None, :]
        else:
            alpha = inputs[:, 0]
        mask = torch.arange(S, device=inputs.device)[None, :] < lengths[:, None]
        backptrs = []
        for t in range(1, S):
            scores = alpha[:, :, None] + trans[None, :, :]  # [B, from, to]
            best, ptr = scores.max(dim=1)
            new_alpha = best + inputs[:, t]
            alpha = torch.where(mask[:, t, None], new_alpha, alpha)
            # frozen rows point to themselves so backtrace stays in place
            ptr = torch.where(mask[:, t, None], ptr,
                              torch.arange(L, device=ptr.device)[None, :].expand_as(ptr))
            backptrs.append(ptr)
        if self.with_start_stop_tag:
            alpha = alpha + self.transitions[:L, stop_idx][None, :]
        best_score, best_tag = alpha.max(dim=1)
        paths = torch.zeros(B, S, dtype=torch.long, device=inputs.device)
        paths[:, S - 1] = best_tag
        cur = best_tag
        for t in range(S - 2, -1, -1):
            cur = backptrs[t].gather(1, cur[:, None]).squeeze(1)
            paths[:, t] = cur
        paths = paths * mask.long()
        return best_score, paths
