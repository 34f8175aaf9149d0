"""Distributed helpers (reference: paddlenlp/trainer/utils/helper.py —
distributed_concat, broadcast_dp_optimizer :233, nested_* utilities)."""
from __future__ import annotations


import torch
import torch.distributed as dist

from ...parallel.topology import get_topology


def distributed_concat(tensor: torch.Tensor, group=None) -> torch.Tensor:
    """All-gather and concatenate along dim 0 (metric/prediction gathering)."""
    if not dist.is_initialized():
        return tensor
    group = group if group is not None else None
    world = dist.get_world_size(group)
    parts = [torch.empty_like(tensor) for _ in range(world)]
    dist.all_gather(parts, tensor.contiguous(), group=group)
    return torch.cat(parts, dim=0)


def nested_concat(tensors, new, dim: int = 0):
    if tensors is None:
        return new
    if isinstance(tensors, (list, tuple)):
        return type(tensors)(nested_concat(t, n, dim) for t, n in zip(tensors, new))
    return torch.cat([tensors, new], dim=dim)


def nested_detach(tensors):
    if isinstance(tensors, (list, tuple)):
        return type(tensors)(nested_detach(t) for t in tensors)
    return tensors.detach() if isinstance(tensors, torch.Tensor) else tensors


def nested_numpify(tensors):
    if isinstance(tensors, (list, tuple)):
        return type(tensors)(nested_numpify(t) for t in tensors)
    return tensors.cpu().numpy() if isinstance(tensors, torch.Tensor) else tensors


def broadcast_dp_optimizer(optimizer_state: dict, group=None) -> dict:
    """Broadcast a loaded optimizer state from dp rank 0 to its replicas
    (reference broadcast_dp_optimizer :233)."""
    topo = get_topology()
    group = group if group is not None else topo.data_parallel_group
    if group is None or not dist.is_initialized():
        return optimizer_state
    obj = [optimizer_state]
    src = dist.get_process_group_ranks(group)[0]
    dist.broadcast_object_list(obj, src=src, group=group)
    return obj[0]


def broadcast_moe_optimizer(optimizer_state: dict, group=None) -> dict:
    """MoE variant: expert states are rank-local (no_sync) and skipped;
    shared states broadcast like dp (reference :233)."""
    topo = get_topology()
    group = group if group is not None else topo.data_parallel_group
    if group is None or not dist.is_initialized():
        return optimizer_state
    shared = {k: v for k, v in optimizer_state.items() if "expert" not in str(k)}
    shared = broadcast_dp_optimizer(shared, group)
    out = dict(optimizer_state)
    out.update(shared)
    return out
