"""Expert parallelism: RCCL all-to-all token dispatch over xGMI.

Reference behavior: SURVEY §2.2 row EP — in the 2024-10 reference, training
EP keeps experts dp-rank-local (no_sync params excluded from dp grad
allreduce, trainer.py:1079-1085) and inference uses a fused_moe op.  The
BASELINE Mixtral config requires real token routing, so this module
implements the standard two-hop dispatch: tokens are sorted by expert,
exchanged with variable splits via a differentiable all-to-all, computed by
the local experts, and returned by the inverse all-to-all.

xGMI note: all-to-all is the natural collective for the point-to-point
fabric — each of the 7 links carries exactly one peer's shard.
"""
from __future__ import annotations

from typing import List

import torch
import torch.distributed as dist



class _AllToAllVar(torch.autograd.Function):
    """Differentiable all_to_all_single with variable splits."""

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        out = x.new_empty(sum(out_splits), *x.shape[1:])
        dist.all_to_all_single(out, x.contiguous(), out_splits, in_splits, group=group)
        return out

    @staticmethod
    def backward(ctx, grad):
        out = grad.new_empty(sum(ctx.in_splits), *grad.shape[1:])
        dist.all_to_all_single(out, grad.contiguous(), ctx.in_splits, ctx.out_splits,
                               group=ctx.group)
        return out, None, None, None


def all_to_all_tokens(x, out_splits: List[int], in_splits: List[int], group):
    return _AllToAllVar.apply(x, out_splits, in_splits, group)


def dispatch_and_combine(
    hidden: torch.Tensor,        # [T, H] flattened tokens (repeated per top-k slot)
    expert_ids: torch.Tensor,    # [T] global expert assignment
    num_experts: int,
    expert_fn,                   # (local_expert_idx, tokens) -> tokens
    group=None,
) -> torch.Tensor:
    """Route each token to its expert (possibly on another rank), apply the
    expert MLP, and return tokens in the original order.

    group=None means no expert parallelism (all experts local)."""
    ep = dist.get_world_size(group) if group is not None else 1
    assert num_experts % ep == 0
    experts_per_rank = num_experts // ep

    # sort tokens by expert
    sort_idx = torch.argsort(expert_ids, stable=True)
    sorted_tokens = hidden[sort_idx]
    counts = torch.bincount(expert_ids, minlength=num_experts)  # [E]

    if ep == 1:
        out_sorted = torch.empty_like(sorted_tokens)
        start = 0
        for e in range(num_experts):
            n = int(counts[e])
            if n:
                out_sorted[start:start + n] = expert_fn(e, sorted_tokens[start:start + n])
            start += n
        out = torch.empty_like(out_sorted)
        out[sort_idx] = out_sorted
        return out

    # exchange token counts: send counts grouped per destination rank
    send_splits = [int(counts[r * experts_per_rank:(r + 1) * experts_per_rank].sum())
                   for r in range(ep)]
    all_counts = [torch.zeros_like(counts) for _ in range(ep)]
    dist.all_gather(all_counts, counts, group=group)
    my_ep_rank = dist.get_rank(group)
    recv_splits = [int(all_counts[r][
        my_ep_rank * experts_per_rank:(my_ep_rank + 1) * experts_per_rank].sum())
        for r in range(ep)]

    recv_tokens = all_to_all_tokens(sorted_tokens, recv_splits, send_splits, group)

    # received tokens arrive grouped by (source rank, local expert); re-sort
    # into local-expert-major order
    local_expert_of_recv = []
    for r in range(ep):
        c = all_counts[r][my_ep_rank * experts_per_rank:(my_ep_rank + 1) * experts_per_rank]
        for le in range(experts_per_rank):
            local_expert_of_recv.append(
                torch.full((int(c[le]),), le, dtype=torch.long, device=hidden.device))
    local_expert_of_recv = (torch.cat(local_expert_of_recv)
                            if local_expert_of_recv
                            else torch.zeros(0, dtype=torch.long, device=hidden.device))
    re_idx = torch.argsort(local_expert_of_recv, stable=True)
    grouped = recv_tokens[re_idx]
    le_counts = torch.bincount(local_expert_of_recv, minlength=experts_per_rank)

    out_grouped = torch.empty_like(grouped)
    start = 0
    for le in range(experts_per_rank):
        n = int(le_counts[le])
        if n:
            out_grouped[start:start + n] = expert_fn(le, grouped[start:start + n])
        start += n

    # undo the local re-sort, send back, undo the original sort
    out_recv = torch.empty_like(out_grouped)
    out_recv[re_idx] = out_grouped
    back = all_to_all_tokens(out_recv, send_splits, recv_splits, group)
    out = torch.empty_like(back)
    out[sort_idx] = back
    return out


def mark_moe_params_no_sync(module: torch.nn.Module):
    """Expert params are dp-excluded (reference trainer.py:1079-1085)."""
    for p in module.parameters():
        p.no_sync = True
