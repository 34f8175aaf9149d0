"""Segment parallel (Ulysses / DeepSpeed-style all-to-all over the sep group).

Reference behavior: paddlenlp/transformers/segment_parallel_utils.py —
ReshardLayer/ReshardQKV all-to-all reshaping [s/sep, b, h] <-> [s, b, h/sep]
(:69-137), applied around attention (llama/modeling.py:811-815), and the
input split on the seq dim (:35, called at trainer.py:972-973).

This framework's layout is [B, S, H(eads), D]: before attention the seq dim
is sharded and heads full; the all-to-all swaps to full seq and sharded
heads so the flash-attention kernel sees the whole sequence.
"""
from __future__ import annotations

import torch
import torch.distributed as dist

from .topology import get_topology


def split_inputs_sequence_dim(inputs: dict, group=None,
                              balanced: bool = False) -> dict:
    """Slice every [B, S, ...] tensor to this rank's seq chunk — contiguous,
    or zigzag (chunks r and 2w-1-r) when `balanced` so causal ring-attention
    work is equal across ranks.

    Reference: trainer.py:972-973 split_inputs_sequence_dim +
    context_parallel_utils.py:43-50 load-balanced chunking."""
    group = group or get_topology().sep_parallel_group
    if group is None:
        return inputs
    rank = dist.get_rank(group)
    world = dist.get_world_size(group)
    out = {}
    for key, v in inputs.items():
        if isinstance(v, torch.Tensor) and v.dim() >= 2 and \
                v.shape[1] % (2 * world if balanced else world) == 0:
            if balanced:
                from .ring_attention import zigzag_split

                out[key] = zigzag_split(v, world, rank, dim=1)
            else:
                out[key] = v.chunk(world, dim=1)[rank].contiguous()
        else:
            out[key] = v
    return out


class _AllToAllSeqToHead(torch.autograd.Function):
    """[B, S/w, H, D] -> [B, S, H/w, D] (and the inverse for backward)."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _a2a_s2h(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _a2a_h2s(grad, ctx.group), None


class _AllToAllHeadToSeq(torch.autograd.Function):
    """[B, S, H/w, D] -> [B, S/w, H, D]."""

    @staticmethod
    def forward(ctx, x, group):
        ctx.group = group
        return _a2a_h2s(x, group)

    @staticmethod
    def backward(ctx, grad):
        return _a2a_s2h(grad, ctx.group), None


def _a2a_s2h(x, group):
    """x: [B, S_local, H, D] -> [B, S_local*w, H/w, D]."""
    w = dist.get_world_size(group)
    B, S, H, D = x.shape
    assert H % w == 0, (H, w)
    # send chunks of heads: peer p receives our seq chunk of its head range
    # reorganize to [w, B, S, H/w, D]
    xs = x.reshape(B, S, w, H // w, D).permute(2, 0, 1, 3, 4).contiguous()
    out = torch.empty_like(xs)
    dist.all_to_all_single(out, xs, group=group)
    # out[p] = peer p's seq chunk of OUR head range -> [B, w*S, H/w, D]
    return out.permute(1, 0, 2, 3, 4).reshape(B, w * S, H // w, D)


def _a2a_h2s(x, group):
    """x: [B, S, H_local, D] -> [B, S/w, H_local*w, D]."""
    w = dist.get_world_size(group)
    B, S, Hl, D = x.shape
    assert S % w == 0
    xs = x.reshape(B, w, S // w, Hl, D).permute(1, 0, 2, 3, 4).contiguous()
    out = torch.empty_like(xs)
    dist.all_to_all_single(out, xs, group=group)
    return out.permute(1, 2, 0, 3, 4).reshape(B, S // w, w * Hl, D)


class ReshardLayer(torch.nn.Module):
    """Ulysses reshard around the attention core (reference ReshardLayer)."""

    def __init__(self, group=None):
        super().__init__()
        self.group = group if group is not None else get_topology().sep_parallel_group

    def seq_to_head(self, x):
        if self.group is None:
            return x
        return _AllToAllSeqToHead.apply(x, self.group)

    def head_to_seq(self, x):
        if self.group is None:
            return x
        return _AllToAllHeadToSeq.apply(x, self.group)
