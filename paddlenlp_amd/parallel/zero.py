"""ZeRO sharding stages 1/2 over RCCL (from-scratch MI355X design).

Reference behavior (SURVEY.md §2.2 row "Sharding stage1/2/3"): stage1 shards
optimizer states (grads all-reduced like DP); stage2 additionally shards
gradients (reduce to the owning rank only); both broadcast updated params
from the owner after the step (reference: fleet group_sharded / DygraphShardingOptimizer
wired at trainer.py:2016-2095).

Design here: parameters are greedily partitioned across the sharding group by
size.  Non-owned params get `grad=None` before optimizer.step() so the
optimizer only touches the local shard (optimizer states therefore only
materialize for owned params = the ZeRO memory saving).  After the step the
owner broadcasts updated param data.

xGMI note: gradient reduction uses per-owner flat buckets
(sharding_comm_buffer_size_MB) so each RCCL call is large enough to hit
per-link bandwidth; reduce-scatter decomposition of all-reduce is what the
ring would do internally anyway, and owner-bucketing gives it to us with
stage-2 memory savings for free.
"""
from __future__ import annotations

from typing import Dict

import torch
import torch.distributed as dist

from ..utils.log import logger


class ZeroShardedEngine:
    def __init__(self, model, optimizer, stage: int, group, bucket_mb: int = 256):
        assert stage in (1, 2), "stage3 lands in a later milestone"
        self.model = model
        self.optimizer = optimizer
        self.stage = stage
        self.group = group
        self.world = dist.get_world_size(group) if group is not None else 1
        self.rank = dist.get_rank(group) if group is not None else 0
        self.group_ranks = dist.get_process_group_ranks(group) if group is not None else [0]
        self.bucket_bytes = bucket_mb * 1024 * 1024

        # greedy size-balanced partition: owner[param] = rank in group
        self.owner: Dict[torch.nn.Parameter, int] = {}
        sizes = [0] * self.world
        params = [p for g in optimizer.param_groups for p in g["params"] if p.requires_grad]
        for p in sorted(params, key=lambda p: -p.numel()):
            r = sizes.index(min(sizes))
            self.owner[p] = r
            sizes[r] += p.numel()
        self.owned_params = [p for p in params if self.owner[p] == self.rank]
        logger.info(
            f"ZeRO stage{stage}: rank {self.rank}/{self.world} owns "
            f"{sum(p.numel() for p in self.owned_params):,} / {sum(sizes):,} params"
        )

    # ------------------------------------------------------------------
    def reduce_gradients_and_step_pre(self):
        """Reduce grads over the sharding group, then drop non-owned grads."""
        if self.group is None:
            return
        params = [p for p in self.owner if p.grad is not None]
        if self.stage == 1:
            # grads replicated: flat all-reduce (average)
            from .data_parallel import fused_allreduce_gradients

            fused_allreduce_gradients(params, self.group, bucket_bytes=self.bucket_bytes)
        else:
            # stage2: reduce each owner's shard to the owner only
            for owner_rank in range(self.world):
                bucket, size = [], 0
                owner_global = self.group_ranks[owner_rank]

                def flush():
                    nonlocal bucket, size
                    if not bucket:
                        return
                    flat = torch._utils._flatten_dense_tensors(bucket)
                    dist.reduce(flat, dst=owner_global, group=self.group)
                    if owner_rank == self.rank:
                        flat.div_(self.world)
                        for g, out in zip(bucket, torch._utils._unflatten_dense_tensors(flat, bucket)):
                            g.copy_(out)
                    bucket, size = [], 0

                for p in params:
                    if self.owner[p] != owner_rank:
                        continue
                    bucket.append(p.grad)
                    size += p.grad.numel() * p.grad.element_size()
                    if size >= self.bucket_bytes:
                        flush()
                flush()
        # drop non-owned grads so the optimizer (and its states) only touch
        # the local shard
        for p in self.owner:
            if self.owner[p] != self.rank:
                p.grad = None

    def step_post(self):
        """Broadcast updated params from their owners (flat-bucketed)."""
        if self.group is None:
            return
        for owner_rank in range(self.world):
            owner_global = self.group_ranks[owner_rank]
            bucket, size = [], 0

            def flush():
                nonlocal bucket, size
                if not bucket:
                    return
                flat = torch._utils._flatten_dense_tensors(bucket)
                dist.broadcast(flat, src=owner_global, group=self.group)
                if owner_rank != self.rank:
                    for p, out in zip(bucket, torch._utils._unflatten_dense_tensors(flat, bucket)):
                        p.copy_(out)
                bucket, size = [], 0

            for p in self.owner:
                if self.owner[p] != owner_rank:
                    continue
                bucket.append(p.data)
                size += p.numel() * p.element_size()
                if size >= self.bucket_bytes:
                    flush()
            flush()
