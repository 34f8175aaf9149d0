"""Sequence parallelism over the tensor-parallel group.

Reference behavior: the framework ops PaddleNLP re-exports at
transformers/__init__.py:37-46 (GatherOp/ScatterOp/AllGatherOp/
ReduceScatterOp, ColumnSequenceParallelLinear / RowSequenceParallelLinear,
register_sequence_parallel_allreduce_hooks) and
transformers/sequence_parallel_utils.py.

Layout note: this framework keeps activations [B, S, H]; sequence parallel
shards dim 1 (S).  The column linear all-gathers the sequence before its
matmul; the row linear reduce-scatters after (Megatron SP pattern).
"""
from __future__ import annotations

import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from .topology import get_topology


def _seq_chunk(x, group):
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    return x.chunk(world, dim=1)[rank].contiguous()


def _all_gather_seq(x, group):
    """[B, S/w, H] -> [B, S, H]."""
    world = dist.get_world_size(group)
    parts = [torch.empty_like(x) for _ in range(world)]
    dist.all_gather(parts, x.contiguous(), group=group)
    return torch.cat(parts, dim=1)


def _reduce_scatter_seq(x, group):
    """[B, S, H] (partial sums) -> [B, S/w, H] (this rank's reduced chunk)."""
    world = dist.get_world_size(group)
    chunks = list(x.chunk(world, dim=1))
    out = torch.empty_like(chunks[0])
    dist.reduce_scatter(out, [c.contiguous() for c in chunks], group=group)
    return out


class _ScatterOp(torch.autograd.Function):
    """fwd: take this rank's seq chunk; bwd: all-gather grads."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _seq_chunk(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _all_gather_seq(grad, ctx.group), None


class _GatherOp(torch.autograd.Function):
    """fwd: all-gather seq; bwd: take this rank's chunk of grads."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _all_gather_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _seq_chunk(grad, ctx.group), None


class _AllGatherOp(torch.autograd.Function):
    """fwd: all-gather seq; bwd: reduce-scatter grads (column-linear input)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _all_gather_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _reduce_scatter_seq(grad, ctx.group), None


class _ReduceScatterOp(torch.autograd.Function):
    """fwd: reduce-scatter seq; bwd: all-gather grads (row-linear output)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _reduce_scatter_seq(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _all_gather_seq(grad, ctx.group), None


def ScatterOp(x, group=None):
    group = group or get_topology().model_parallel_group
    if group is None:
        return x
    return _ScatterOp.apply(x, group)


def GatherOp(x, group=None):
    group = group or get_topology().model_parallel_group
    if group is None:
        return x
    return _GatherOp.apply(x, group)


def AllGatherOp(x, group=None):
    group = group or get_topology().model_parallel_group
    if group is None:
        return x
    return _AllGatherOp.apply(x, group)


def ReduceScatterOp(x, group=None):
    group = group or get_topology().model_parallel_group
    if group is None:
        return x
    return _ReduceScatterOp.apply(x, group)


class ColumnSequenceParallelLinear(nn.Module):
    """All-gather the sequence-sharded input, then column-parallel matmul.

    Input [B, S/mp, H] -> output [B, S, out/mp]."""

    def __init__(self, in_features, out_features, bias=False, gather_output=False, group=None):
        super().__init__()
        self.group = group if group is not None else get_topology().model_parallel_group
        world = dist.get_world_size(self.group) if self.group is not None else 1
        assert out_features % world == 0
        self.weight = nn.Parameter(torch.empty(out_features // world, in_features))
        self.weight.is_column_parallel = True
        self.bias = nn.Parameter(torch.zeros(out_features // world)) if bias else None
        if self.bias is not None:
            self.bias.is_column_parallel = True
        nn.init.normal_(self.weight, std=0.02)
        assert not gather_output, "SP column linear keeps sharded outputs"

    def forward(self, x):
        if self.group is not None:
            x = _AllGatherOp.apply(x, self.group)
        return F.linear(x, self.weight, self.bias)


class RowSequenceParallelLinear(nn.Module):
    """Row-parallel matmul, then reduce-scatter the sequence.

    Input [B, S, in/mp] -> output [B, S/mp, out]."""

    def __init__(self, in_features, out_features, bias=False, input_is_parallel=True, group=None):
        super().__init__()
        self.group = group if group is not None else get_topology().model_parallel_group
        world = dist.get_world_size(self.group) if self.group is not None else 1
        assert in_features % world == 0
        self.weight = nn.Parameter(torch.empty(out_features, in_features // world))
        self.weight.is_row_parallel = True
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        y = F.linear(x, self.weight)
        if self.group is not None:
            y = _ReduceScatterOp.apply(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


def mark_as_sequence_parallel_parameter(param: torch.nn.Parameter):
    param.sequence_parallel = True


def register_sequence_parallel_allreduce_hooks(model: nn.Module, group=None):
    """Grad all-reduce over the mp group for params that see sequence-sharded
    activations but are NOT tensor-parallel themselves (norm weights, biases).

    Reference: register_sequence_parallel_allreduce_hooks
    (llm/run_pretrain.py:503).  Call after backward (the trainer does this in
    optimizer_step) or rely on the registered autograd hooks.
    """
    group = group or get_topology().model_parallel_group
    if group is None:
        return

    def make_hook(p):
        def hook(grad):
            grad = grad.contiguous()
            dist.all_reduce(grad, group=group)
            return grad
        return hook

    for name, p in model.named_parameters():
        if getattr(p, "sequence_parallel", False):
            p.register_hook(make_hook(p))
