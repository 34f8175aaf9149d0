"""Tensor-parallel layers over RCCL (torch.distributed) — built from scratch.

Reference behavior: the fleet layers PaddleNLP aliases in
paddlenlp/transformers/linear_utils.py:33-38 (ColumnParallelLinear /
RowParallelLinear / VocabParallelEmbedding) and the helpers
parallel_matmul (llama/modeling.py:176-204) and ParallelCrossEntropy
(llama/modeling.py:1795).

Collective pattern (Megatron-style):
  ColumnParallelLinear: X @ [W1 | W2]  — identity fwd / all-reduce bwd on input
  RowParallelLinear:    [X1 X2] @ [W1; W2] — all-reduce fwd / identity bwd
On MI355X the mp group is xGMI-adjacent ranks, so these all-reduces run over
direct links.
"""
from __future__ import annotations


import torch
import torch.distributed as dist
import torch.nn as nn
import torch.nn.functional as F

from .topology import get_topology


# ---------------------------------------------------------------------------
# autograd-aware collectives
# ---------------------------------------------------------------------------
class _CopyToModelParallel(torch.autograd.Function):
    """Identity fwd; all-reduce grad bwd (input of ColumnParallelLinear)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return x

    @staticmethod
    def backward(ctx, grad):
        if ctx.group is not None:
            grad = grad.contiguous()
            dist.all_reduce(grad, group=ctx.group)
        return grad, None


class _ReduceFromModelParallel(torch.autograd.Function):
    """All-reduce fwd; identity bwd (output of RowParallelLinear)."""

    @staticmethod
    def forward(ctx, x, group):
        if group is not None:
            x = x.contiguous()
            dist.all_reduce(x, group=group)
        return x

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class _GatherFromModelParallel(torch.autograd.Function):
    """All-gather along last dim fwd; slice bwd (gather sharded logits)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        if group is None:
            return x
        world = dist.get_world_size(group)
        ctx.world = world
        ctx.rank = dist.get_rank(group)
        parts = [torch.empty_like(x) for _ in range(world)]
        dist.all_gather(parts, x.contiguous(), group=group)
        return torch.cat(parts, dim=-1)

    @staticmethod
    def backward(ctx, grad):
        if ctx.group is None:
            return grad, None
        dim = grad.shape[-1] // ctx.world
        return grad[..., ctx.rank * dim:(ctx.rank + 1) * dim].contiguous(), None


def copy_to_model_parallel(x, group):
    return _CopyToModelParallel.apply(x, group)


def reduce_from_model_parallel(x, group):
    return _ReduceFromModelParallel.apply(x, group)


def gather_from_model_parallel(x, group):
    return _GatherFromModelParallel.apply(x, group)


# ---------------------------------------------------------------------------
# layers
# ---------------------------------------------------------------------------
class ColumnParallelLinear(nn.Module):
    """Y_local = X @ W_local^T (+ b_local); W is split on output dim."""

    def __init__(self, in_features, out_features, bias=False, gather_output=False, group=None):
        super().__init__()
        self.group = group if group is not None else get_topology().model_parallel_group
        world = dist.get_world_size(self.group) if self.group is not None else 1
        assert out_features % world == 0, (out_features, world)
        self.in_features = in_features
        self.out_features_per_partition = out_features // world
        self.gather_output = gather_output
        self.weight = nn.Parameter(
            torch.empty(self.out_features_per_partition, in_features)
        )
        self.weight.is_column_parallel = True
        self.bias = None
        if bias:
            self.bias = nn.Parameter(torch.zeros(self.out_features_per_partition))
            self.bias.is_column_parallel = True
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        x = copy_to_model_parallel(x, self.group)
        y = F.linear(x, self.weight, self.bias)
        if self.gather_output:
            y = gather_from_model_parallel(y, self.group)
        return y


class RowParallelLinear(nn.Module):
    """Y = sum_ranks(X_local @ W_local^T) + b; W split on input dim."""

    def __init__(self, in_features, out_features, bias=False, input_is_parallel=True, group=None):
        super().__init__()
        self.group = group if group is not None else get_topology().model_parallel_group
        world = dist.get_world_size(self.group) if self.group is not None else 1
        assert in_features % world == 0, (in_features, world)
        self.in_features_per_partition = in_features // world
        self.out_features = out_features
        self.input_is_parallel = input_is_parallel
        self.weight = nn.Parameter(
            torch.empty(out_features, self.in_features_per_partition)
        )
        self.weight.is_row_parallel = True
        self.bias = nn.Parameter(torch.zeros(out_features)) if bias else None
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, x):
        y = F.linear(x, self.weight)
        y = reduce_from_model_parallel(y, self.group)
        if self.bias is not None:
            y = y + self.bias
        return y


class VocabParallelEmbedding(nn.Module):
    """Embedding with the vocab dim split across mp ranks; all-reduce output.

    Reference: llama/modeling.py:1459."""

    def __init__(self, num_embeddings, embedding_dim, group=None):
        super().__init__()
        self.group = group if group is not None else get_topology().model_parallel_group
        world = dist.get_world_size(self.group) if self.group is not None else 1
        rank = dist.get_rank(self.group) if self.group is not None else 0
        assert num_embeddings % world == 0, (num_embeddings, world)
        self.num_embeddings = num_embeddings
        self.per_partition = num_embeddings // world
        self.vocab_start = rank * self.per_partition
        self.vocab_end = self.vocab_start + self.per_partition
        self.weight = nn.Parameter(torch.empty(self.per_partition, embedding_dim))
        self.weight.is_column_parallel = True  # split on vocab(out) dim
        nn.init.normal_(self.weight, std=0.02)

    def forward(self, input_ids):
        if self.group is None:
            return F.embedding(input_ids, self.weight)
        mask = (input_ids < self.vocab_start) | (input_ids >= self.vocab_end)
        masked = input_ids.clamp(self.vocab_start, self.vocab_end - 1) - self.vocab_start
        out = F.embedding(masked, self.weight)
        out = out.masked_fill(mask.unsqueeze(-1), 0.0)
        return reduce_from_model_parallel(out, self.group)


def parallel_matmul(x: torch.Tensor, weight: torch.Tensor, tensor_parallel_output: bool = True,
                    transpose_y: bool = True, group=None):
    """LM-head matmul with a vocab-sharded weight.

    Reference: llama/modeling.py:176-204.  If tensor_parallel_output, each
    rank keeps its vocab shard of the logits (feed ParallelCrossEntropy);
    else all-gather the full logits.
    """
    group = group if group is not None else get_topology().model_parallel_group
    if group is not None:
        x = copy_to_model_parallel(x, group)
    logits = x @ (weight.t() if transpose_y else weight)
    if group is not None and not tensor_parallel_output:
        logits = gather_from_model_parallel(logits, group)
    return logits


class _ParallelCrossEntropy(torch.autograd.Function):
    """Cross-entropy on vocab-sharded logits without gathering them.

    Per-rank: local max + local sum-exp + local target logit, then three
    cheap all-reduces.  Returns per-token loss [N].
    """

    @staticmethod
    def forward(ctx, logits, labels, group, ignore_index):
        world = dist.get_world_size(group) if group is not None else 1
        rank = dist.get_rank(group) if group is not None else 0
        n, v_local = logits.shape
        vocab_start = rank * v_local

        logits32 = logits.float()
        local_max = logits32.max(dim=-1).values
        if group is not None:
            gmax = local_max.clone()
            dist.all_reduce(gmax, op=dist.ReduceOp.MAX, group=group)
        else:
            gmax = local_max
        shifted = logits32 - gmax.unsqueeze(-1)
        exp = shifted.exp()
        sum_exp = exp.sum(-1)
        if group is not None:
            dist.all_reduce(sum_exp, group=group)

        # gather target logit where the label lives on this rank
        labels_local = labels - vocab_start
        in_range = (labels_local >= 0) & (labels_local < v_local)
        safe = labels_local.clamp(0, v_local - 1)
        tgt = shifted.gather(-1, safe.unsqueeze(-1)).squeeze(-1)
        tgt = torch.where(in_range, tgt, torch.zeros_like(tgt))
        if group is not None:
            dist.all_reduce(tgt, group=group)

        lse = sum_exp.log()
        loss = lse - tgt
        valid = labels != ignore_index
        loss = torch.where(valid, loss, torch.zeros_like(loss))

        softmax = exp / sum_exp.unsqueeze(-1)
        ctx.save_for_backward(softmax, labels_local, in_range, valid)
        ctx.dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, dloss):
        softmax, labels_local, in_range, valid = ctx.saved_tensors
        grad = softmax
        idx = labels_local.clamp(0, softmax.shape[-1] - 1)
        one_hot = torch.zeros_like(grad)
        one_hot.scatter_(-1, idx.unsqueeze(-1), in_range.to(grad.dtype).unsqueeze(-1))
        grad = grad - one_hot
        grad = grad * (dloss * valid.to(grad.dtype)).unsqueeze(-1)
        return grad.to(ctx.dtype), None, None, None


class ParallelCrossEntropy(nn.Module):
    def __init__(self, group=None, ignore_index: int = -100):
        super().__init__()
        self.group = group if group is not None else get_topology().model_parallel_group
        self.ignore_index = ignore_index

    def forward(self, logits, labels):
        """logits: [N, V_local] (sharded) or [N, V] when group is None."""
        orig_shape = labels.shape
        logits = logits.reshape(-1, logits.shape[-1])
        labels = labels.reshape(-1)
        loss = _ParallelCrossEntropy.apply(logits, labels, self.group, self.ignore_index)
        return loss.reshape(orig_shape)
