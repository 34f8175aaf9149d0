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
    """Differentiable all_to_all_single with variable splits.  An optional
    `overlap_fn` runs between issuing the exchange and waiting on it, so
    independent compute (e.g. a shared expert) hides the xGMI hop."""

    _overlap_fn = None

    @staticmethod
    def forward(ctx, x, out_splits, in_splits, group):
        ctx.group = group
        ctx.out_splits = out_splits
        ctx.in_splits = in_splits
        out = x.new_empty(sum(out_splits), *x.shape[1:])
        fn = _AllToAllVar._overlap_fn
        if fn is not None:
            _AllToAllVar._overlap_fn = None
            work = dist.all_to_all_single(out, x.contiguous(), out_splits,
                                          in_splits, group=group, async_op=True)
            fn()
            work.wait()
        else:
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
    expert_fn=None,              # (local_expert_idx, tokens) -> tokens
    group=None,
    grouped_fn=None,             # (expert-sorted tokens, per-expert counts) -> tokens
    overlap_fn=None,             # runs under the dispatch all-to-all (shared expert)
) -> torch.Tensor:
    """Route each token to its expert (possibly on another rank), apply the
    expert MLP, and return tokens in the original order.

    group=None means no expert parallelism (all experts local).
    `grouped_fn`, when given, receives ALL local tokens sorted by expert
    plus the per-local-expert counts — one batched GEMM instead of a
    per-expert loop (reference fused_moe,
    fused_transformer_layers.py:951-1008)."""
    ep = dist.get_world_size(group) if group is not None else 1
    assert num_experts % ep == 0
    experts_per_rank = num_experts // ep

    # sort tokens by expert
    sort_idx = torch.argsort(expert_ids, stable=True)
    sorted_tokens = hidden[sort_idx]
    counts = torch.bincount(expert_ids, minlength=num_experts)  # [E]

    if ep == 1:
        if overlap_fn is not None:
            overlap_fn()
        if grouped_fn is not None:
            out_sorted = grouped_fn(sorted_tokens, counts)
        else:
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

    if overlap_fn is not None:
        _AllToAllVar._overlap_fn = overlap_fn
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

    if grouped_fn is not None:
        out_grouped = grouped_fn(grouped, le_counts)
    else:
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


class GroupedExperts(torch.nn.Module):
    """Stacked expert-MLP weights computed with ONE batched GEMM per
    projection over a capacity-padded token tensor — the MI355X
    replacement for the dense per-expert loop (reference fused_moe op,
    fused_transformer_layers.py:951-1008).  hipBLASLt's batched bf16 GEMM
    keeps the MFMA pipes full where E sequential skinny GEMMs are
    launch-bound.

    State-dict keys stay per-expert ("{e}.w1.weight" etc., matching the
    nn.ModuleList layout) so checkpoints and TP/HF conversion mappings
    are unchanged."""

    def __init__(self, n_experts: int, hidden: int, intermediate: int,
                 names=("w1", "w3", "w2")):
        super().__init__()
        self.n_experts = n_experts
        self.hidden = hidden
        self.intermediate = intermediate
        self.names = names   # (gate, up, down) state-dict names
        # bmm layouts: x [E, C, H] @ w1 [E, H, I]
        self.w1 = torch.nn.Parameter(torch.empty(n_experts, hidden, intermediate))
        self.w3 = torch.nn.Parameter(torch.empty(n_experts, hidden, intermediate))
        self.w2 = torch.nn.Parameter(torch.empty(n_experts, intermediate, hidden))
        for w, fan_in in ((self.w1, hidden), (self.w3, hidden), (self.w2, intermediate)):
            bound = 1.0 / (fan_in ** 0.5)
            torch.nn.init.uniform_(w, -bound, bound)

    def __len__(self):
        return self.n_experts

    def __getitem__(self, e):
        """Single-expert callable view (ModuleList compatibility)."""
        def f(x):
            from ..ops import functional as ops

            gu = torch.cat([x @ self.w1[e].to(x.dtype),
                            x @ self.w3[e].to(x.dtype)], dim=-1)
            return ops.swiglu(gu) @ self.w2[e].to(x.dtype)
        return f

    def forward_grouped(self, x: torch.Tensor, counts: torch.Tensor) -> torch.Tensor:
        """x: [N, H] tokens sorted by expert; counts: [E] per-expert."""
        from ..ops import functional as ops

        N, H = x.shape
        E = self.n_experts
        if N == 0:
            return x
        C = int(counts.max().item())
        if C == 0:
            return x
        dev = x.device
        offsets = torch.cumsum(counts, 0) - counts
        tok_expert = torch.repeat_interleave(
            torch.arange(E, device=dev), counts.to(dev))
        pos = torch.arange(N, device=dev) - offsets.to(dev)[tok_expert]
        idx = tok_expert * C + pos
        xp = x.new_zeros(E * C, H)
        xp = xp.index_copy(0, idx, x).view(E, C, H)
        w1 = self.w1.to(x.dtype)
        w3 = self.w3.to(x.dtype)
        w2 = self.w2.to(x.dtype)
        gu = torch.cat([torch.bmm(xp, w1), torch.bmm(xp, w3)], dim=-1)
        act = ops.swiglu(gu)
        yp = torch.bmm(act, w2)
        return yp.reshape(E * C, H).index_select(0, idx)

    # ---- per-expert state-dict interface (ModuleList-compatible) ----
    def _save_to_state_dict(self, destination, prefix, keep_vars):
        g, u, d = self.names
        for e in range(self.n_experts):
            destination[f"{prefix}{e}.{g}.weight"] = self.w1[e].t().contiguous()
            destination[f"{prefix}{e}.{u}.weight"] = self.w3[e].t().contiguous()
            destination[f"{prefix}{e}.{d}.weight"] = self.w2[e].t().contiguous()

    def _load_from_state_dict(self, state_dict, prefix, local_metadata, strict,
                              missing_keys, unexpected_keys, error_msgs):
        g, u, d = self.names
        with torch.no_grad():
            for e in range(self.n_experts):
                for name, stacked in ((g, self.w1), (u, self.w3), (d, self.w2)):
                    key = f"{prefix}{e}.{name}.weight"
                    if key in state_dict:
                        t = state_dict[key]
                        stacked[e].copy_(t.t().to(stacked.dtype))
                    elif strict:
                        missing_keys.append(key)
