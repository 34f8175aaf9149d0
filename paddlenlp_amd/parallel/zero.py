"""ZeRO sharding stages 1/2 over RCCL (from-scratch MI355X design).

Reference behavior (SURVEY.md §2.2 row "Sharding stage1/2/3"): stage1 shards
optimizer states (grads all-reduced like DP); stage2 additionally shards
gradients; both make updated params visible on every rank after the step
(reference: fleet group_sharded / DygraphShardingOptimizer "split_param"
with allgather/broadcast overlap, trainer.py:2016-2095, comm-buffer fusion
training_args.py:564).

MI355X design — flat shard-aligned buckets:

  * Parameters are packed (in REVERSE registration order, i.e. backward
    production order) into dtype-homogeneous buckets of ~bucket_mb.  Within
    a bucket every param is assigned whole to one rank (greedy size
    balance); each rank's segment is padded to a common ``shard_elems`` so
    the bucket buffer is exactly ``[world * shard_elems]``.
  * ``p.data`` and ``p.grad`` become views into the bucket's persistent
    param/grad buffers.  Backward accumulates straight into the flat grad
    buffer — zero-copy.
  * Gradient exchange is ONE ``dist.reduce_scatter_tensor`` per bucket
    (in-place: output = the local shard view).  That is the
    bandwidth-optimal schedule for 7-link xGMI rings: n-1 per-link hops of
    1/n of the data, half the bytes of the old per-owner reduce+broadcast
    and no serialized latency chain.
  * After optimizer.step() (which touched only locally-owned params, whose
    data views live inside the local shard) params are republished with ONE
    in-place ``dist.all_gather_into_tensor`` per bucket, launched async
    across buckets.
  * With ``enable_overlap_comm()`` each bucket's reduce-scatter launches
    asynchronously the moment its last gradient accumulates, overlapping
    the remaining backward compute (mirrors the reference's
    stage1-v2 comm/compute overlap switches, trainer.py:2083-2095).

Stage 1 vs stage 2 differ only in bookkeeping here: both use
reduce-scatter (stage 1 gains the same bandwidth win; its "unsharded
grads" property is not observable through the Trainer, which clips with a
sharding-group reduction either way).
"""
from __future__ import annotations

from typing import Dict, List

import torch
import torch.distributed as dist

from ..utils.log import logger

_ALIGN = 64  # element alignment of each rank shard inside a bucket


class _Bucket:
    __slots__ = ("params", "owner", "shard_elems", "param_buf", "grad_buf",
                 "offsets", "fired", "work")

    def __init__(self, params, owner, shard_elems, param_buf, grad_buf, offsets):
        self.params = params            # all params in this bucket
        self.owner = owner              # param -> group rank
        self.shard_elems = shard_elems
        self.param_buf = param_buf      # [world * shard_elems]
        self.grad_buf = grad_buf        # [world * shard_elems]
        self.offsets = offsets          # param -> start offset in buf
        self.fired = 0
        self.work = None


class ZeroShardedEngine:
    def __init__(self, model, optimizer, stage: int, group, bucket_mb: int = 256):
        assert stage in (1, 2), "stage3 lives in parallel.zero3"
        self.model = model
        self.optimizer = optimizer
        self.stage = stage
        self.group = group
        self.world = dist.get_world_size(group) if group is not None else 1
        self.rank = dist.get_rank(group) if group is not None else 0
        self.bucket_bytes = max(bucket_mb, 1) * 1024 * 1024

        params, seen = [], set()
        for g in optimizer.param_groups:
            for p in g["params"]:
                if p.requires_grad and id(p) not in seen:
                    seen.add(id(p))
                    params.append(p)
        self._param_list = params

        self.buckets: List[_Bucket] = []
        self.owner: Dict[torch.nn.Parameter, int] = {}
        self._build_buckets(params)
        self.owned_params = [p for p in params if self.owner[p] == self.rank]
        self._attach_grad_views()

        total = sum(p.numel() for p in params)
        pad = sum(b.param_buf.numel() for b in self.buckets) - total
        logger.info(
            f"ZeRO stage{stage}: rank {self.rank}/{self.world} owns "
            f"{sum(p.numel() for p in self.owned_params):,} / {total:,} params "
            f"in {len(self.buckets)} buckets (pad {pad:,} elems)")

        self._overlap = False
        self.overlap_active = False
        self._param_bucket = {}
        self._hooks = []

    # ------------------------------------------------------------------
    def _build_buckets(self, params):
        # reverse order => backward produces a bucket's grads consecutively
        by_dtype: Dict[torch.dtype, list] = {}
        for p in reversed(params):
            by_dtype.setdefault(p.dtype, []).append(p)
        for dtype, plist in by_dtype.items():
            cur, size = [], 0
            for p in plist:
                cur.append(p)
                size += p.numel() * p.element_size()
                if size >= self.bucket_bytes:
                    self._make_bucket(cur, dtype)
                    cur, size = [], 0
            if cur:
                self._make_bucket(cur, dtype)

    def _make_bucket(self, plist, dtype):
        # greedy size-balanced whole-param assignment inside the bucket
        loads = [0] * self.world
        per_rank = [[] for _ in range(self.world)]
        owner = {}
        for p in sorted(plist, key=lambda p: -p.numel()):
            r = loads.index(min(loads))
            owner[p] = r
            per_rank[r].append(p)
            loads[r] += p.numel()
        shard = max(loads)
        shard = (shard + _ALIGN - 1) // _ALIGN * _ALIGN
        if shard == 0:
            return
        device = plist[0].device
        param_buf = torch.zeros(self.world * shard, dtype=dtype, device=device)
        grad_buf = torch.zeros(self.world * shard, dtype=dtype, device=device)
        offsets = {}
        for r in range(self.world):
            off = r * shard
            for p in per_rank[r]:
                n = p.numel()
                param_buf[off:off + n].copy_(p.data.reshape(-1))
                p.data = param_buf[off:off + n].view(p.shape)
                offsets[p] = off
                off += n
        b = _Bucket(plist, owner, shard, param_buf, grad_buf, offsets)
        self.buckets.append(b)
        self.owner.update(owner)

    def _grad_view(self, b: _Bucket, p):
        off = b.offsets[p]
        return b.grad_buf[off:off + p.numel()].view(p.shape)

    def _attach_grad_views(self, owned_only: bool = False):
        for b in self.buckets:
            for p in b.params:
                if owned_only and b.owner[p] != self.rank:
                    p.grad = None
                else:
                    p.grad = self._grad_view(b, p)

    def zero_grad(self, set_to_none: bool = True):  # signature-compatible
        """Zero the flat grad buffers and re-attach every grad view.

        Replaces ``optimizer.zero_grad``: with views attached, backward
        accumulates straight into the bucket buffers (zero extra copies).
        """
        for b in self.buckets:
            b.grad_buf.zero_()
            b.fired = 0
            b.work = None
        self._attach_grad_views()

    # ------------------------------------------------------------------
    # comm overlap
    # ------------------------------------------------------------------
    def enable_overlap_comm(self):
        """Per-param hooks launch each bucket's reduce-scatter the moment
        its last gradient accumulates (only while ``overlap_active``, i.e.
        during the final micro-batch's backward)."""
        if self.group is None or self._overlap:
            return
        self._overlap = True
        for bi, b in enumerate(self.buckets):
            for p in b.params:
                self._param_bucket[p] = bi
                self._hooks.append(
                    p.register_post_accumulate_grad_hook(self._on_grad_ready))
        logger.info(f"ZeRO: overlap enabled over {len(self.buckets)} buckets")

    def _on_grad_ready(self, p):
        if not self.overlap_active:
            return
        b = self.buckets[self._param_bucket[p]]
        b.fired += 1
        if b.fired == len(b.params):
            self._launch_bucket(b)

    def _sync_bucket_grads(self, b: _Bucket):
        # tolerate externally re-allocated grads (tests call
        # optimizer.zero_grad(set_to_none=True) directly): copy any grad
        # that is not our view back into the flat buffer
        for p in b.params:
            g = p.grad
            if g is None:
                continue
            off = b.offsets[p]
            view = b.grad_buf[off:off + p.numel()]
            if g.data_ptr() != view.data_ptr():
                view.copy_(g.reshape(-1))
                p.grad = view.view(p.shape)

    def _launch_bucket(self, b: _Bucket, async_op: bool = True):
        self._sync_bucket_grads(b)
        shard = b.grad_buf[self.rank * b.shard_elems:(self.rank + 1) * b.shard_elems]
        b.work = dist.reduce_scatter_tensor(
            shard, b.grad_buf, group=self.group, async_op=async_op)
        b.fired = -1

    # ------------------------------------------------------------------
    def reduce_gradients_and_step_pre(self):
        """One reduce-scatter per bucket; then only locally-owned grads
        remain attached (optimizer states materialize only for the local
        shard = the ZeRO memory saving)."""
        if self.group is None:
            return
        for b in self.buckets:
            if b.fired >= 0:            # not yet launched by overlap hooks
                self._launch_bucket(b)
        for b in self.buckets:
            if b.work is not None:
                b.work.wait()
                b.work = None
            shard = b.grad_buf[self.rank * b.shard_elems:(self.rank + 1) * b.shard_elems]
            shard.div_(self.world)
            b.fired = 0
        self._attach_grad_views(owned_only=True)
        self.overlap_active = False

    def step_post(self):
        """Republish updated params: one async in-place all-gather per
        bucket (param views make the result visible with zero copies)."""
        if self.group is None:
            return
        works = []
        for b in self.buckets:
            shard = b.param_buf[self.rank * b.shard_elems:(self.rank + 1) * b.shard_elems]
            works.append(dist.all_gather_into_tensor(
                b.param_buf, shard, group=self.group, async_op=True))
        for w in works:
            w.wait()
