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

        # ---- optional comm/compute overlap (enable_overlap_comm) ----
        # Buckets in REVERSE parameter order (the backward pass produces
        # grads last-layer-first) so each bucket's reduce launches as soon
        # as its grads exist, overlapping with the rest of backward.
        self._overlap = False
        self.overlap_active = False
        self._param_list = params
        self._buckets = []          # dicts: params, owner_rank, fired
        self._param_bucket = {}
        self._pending = []          # (work, flat, grads, owner_rank)
        self._hooks = []

    # ------------------------------------------------------------------
    # comm overlap
    # ------------------------------------------------------------------
    def enable_overlap_comm(self):
        """Register per-param hooks that reduce each owner-bucket
        asynchronously the moment its last gradient accumulates (only while
        `overlap_active`, i.e. during the final micro-batch's backward)."""
        if self.group is None or self._overlap:
            return
        self._overlap = True
        per_owner = {r: [] for r in range(self.world)}
        for p in reversed(self._param_list):
            per_owner[self.owner[p]].append(p)
        for owner_rank, plist in per_owner.items():
            bucket, size = [], 0
            for p in plist:
                bucket.append(p)
                size += p.numel() * p.element_size()
                if size >= self.bucket_bytes:
                    self._buckets.append(
                        {"params": bucket, "owner_rank": owner_rank, "fired": 0})
                    bucket, size = [], 0
            if bucket:
                self._buckets.append(
                    {"params": bucket, "owner_rank": owner_rank, "fired": 0})
        for bi, b in enumerate(self._buckets):
            for p in b["params"]:
                self._param_bucket[p] = bi
                self._hooks.append(p.register_post_accumulate_grad_hook(
                    self._on_grad_ready))
        logger.info(f"ZeRO: overlap enabled over {len(self._buckets)} buckets")

    def _on_grad_ready(self, p):
        if not self.overlap_active:
            return
        b = self._buckets[self._param_bucket[p]]
        b["fired"] += 1
        if b["fired"] == len(b["params"]):
            self._launch_bucket(b)

    def _launch_bucket(self, b):
        grads = [p.grad for p in b["params"] if p.grad is not None]
        if not grads:
            b["fired"] = -1  # mark done
            return
        flat = torch._utils._flatten_dense_tensors(grads)
        if self.stage == 1:
            work = dist.all_reduce(flat, group=self.group, async_op=True)
        else:
            work = dist.reduce(flat, dst=self.group_ranks[b["owner_rank"]],
                               group=self.group, async_op=True)
        self._pending.append((work, flat, grads, b["owner_rank"]))
        b["fired"] = -1

    def _finish_overlap(self):
        # launch any bucket that never completed (params without grads)
        for b in self._buckets:
            if b["fired"] >= 0:
                self._launch_bucket(b)
        for work, flat, grads, owner_rank in self._pending:
            work.wait()
            if self.stage == 1 or owner_rank == self.rank:
                flat.div_(self.world)
                for g, out in zip(grads,
                                  torch._utils._unflatten_dense_tensors(flat, grads)):
                    g.copy_(out)
        self._pending.clear()
        for b in self._buckets:
            b["fired"] = 0
        self.overlap_active = False

    # ------------------------------------------------------------------
    def reduce_gradients_and_step_pre(self):
        """Reduce grads over the sharding group, then drop non-owned grads."""
        if self.group is None:
            return
        if self._overlap and (self.overlap_active or self._pending):
            self._finish_overlap()
            for p in self.owner:
                if self.owner[p] != self.rank:
                    p.grad = None
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
