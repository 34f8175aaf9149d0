"""DistDataLoader: read data only on dp-source ranks, broadcast to mp/pp peers.

Reference behavior: paddlenlp/data/dist_dataloader.py:41 — only ranks with
need_data (mp rank 0, pp first/last stage...) construct the real loader and
broadcast batches to their mp/pp peers (:150-213).  v1 broadcasts to the mp
group (tensor-parallel peers always need identical batches); pp stages all
receive the batch too (first stage uses input_ids, last uses labels).
"""
from __future__ import annotations

from typing import Iterator

import torch
import torch.distributed as dist

from ..parallel.topology import get_topology


class DistDataLoader:
    def __init__(self, dataset, batch_sampler=None, collate_fn=None,
                 num_workers: int = 0, pin_memory: bool = False, topology=None):
        self.topo = topology or get_topology()
        self._need_data = (
            self.topo.get_rank_in("mp") == 0 and self.topo.get_rank_in("pp") == 0
            and self.topo.get_rank_in("sep") == 0
        )
        self._loader = None
        self._length = None
        if self._need_data:
            self._loader = torch.utils.data.DataLoader(
                dataset, batch_sampler=batch_sampler, collate_fn=collate_fn,
                num_workers=num_workers, pin_memory=pin_memory,
            )
            self._length = len(self._loader) if batch_sampler is not None else None
        # everyone needs the length for the training loop
        if dist.is_initialized():
            t = torch.tensor([self._length if self._length is not None else -1],
                             dtype=torch.int64)
            src_groups = [("mp", self.topo.model_parallel_group),
                          ("pp", self.topo.pipe_parallel_group),
                          ("sep", self.topo.sep_parallel_group)]
            for axis, group in src_groups:
                if group is not None:
                    src = dist.get_process_group_ranks(group)[0]
                    dist.broadcast(t, src=src, group=group)
            self._length = int(t.item()) if t.item() >= 0 else None

    def __len__(self):
        return self._length if self._length is not None else 0

    def _broadcast_batch(self, batch):
        """Broadcast a dict of tensors from the data-reading rank over the
        mp, pp and sep groups (reference _broadcast_data :150-213)."""
        groups = [g for g in (self.topo.model_parallel_group,
                              self.topo.pipe_parallel_group,
                              self.topo.sep_parallel_group) if g is not None]
        for group in groups:
            src = dist.get_process_group_ranks(group)[0]
            obj = [batch if batch is not None else None]
            dist.broadcast_object_list(obj, src=src, group=group)
            batch = obj[0]
        return batch

    def __iter__(self) -> Iterator:
        if not dist.is_initialized() or (
            self.topo.mp_degree == 1 and self.topo.pp_degree == 1 and self.topo.sep_degree == 1
        ):
            yield from self._loader
            return
        if self._need_data:
            for batch in self._loader:
                yield self._broadcast_batch(batch)
        else:
            for _ in range(len(self)):
                yield self._broadcast_batch(None)
