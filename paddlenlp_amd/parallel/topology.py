"""Hybrid-parallel rank topology over torch.distributed (RCCL on ROCm).

Reference behavior: paddle fleet's HybridCommunicateGroup as configured by
paddlenlp/trainer/training_args.py:1110-1405 — an ordered rank grid over the
axes ["dp", "pp", "sharding", "sep", "mp"] (dp slowest-varying, mp
fastest-varying so tensor-parallel peers are xGMI neighbors), with one
process group per axis.

MI355X notes: the "nccl" backend IS RCCL on ROCm.  mp (tensor parallel) ranks
are adjacent so TP collectives run over direct xGMI links; sep/context
parallel shares one axis (reference training_args.py:1284-1286).
"""
from __future__ import annotations

import os
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

from ..utils.log import logger

_AXES = ["dp", "pp", "sharding", "sep", "mp"]

_TOPOLOGY: Optional["Topology"] = None


@dataclass
class Topology:
    """Rank grid + per-axis process groups."""

    dp_degree: int = 1
    pp_degree: int = 1
    sharding_degree: int = 1
    sep_degree: int = 1
    mp_degree: int = 1
    rank: int = 0
    world_size: int = 1
    # per-axis process group and this rank's coordinates
    groups: Dict[str, Optional[dist.ProcessGroup]] = field(default_factory=dict)
    group_ranks: Dict[str, List[int]] = field(default_factory=dict)
    coords: Dict[str, int] = field(default_factory=dict)

    @property
    def degrees(self) -> Dict[str, int]:
        return {
            "dp": self.dp_degree,
            "pp": self.pp_degree,
            "sharding": self.sharding_degree,
            "sep": self.sep_degree,
            "mp": self.mp_degree,
        }

    # axis accessors -----------------------------------------------------
    def get_group(self, axis: str) -> Optional[dist.ProcessGroup]:
        return self.groups.get(axis)

    def get_rank_in(self, axis: str) -> int:
        return self.coords.get(axis, 0)

    def get_degree(self, axis: str) -> int:
        return self.degrees[axis]

    # convenience names mirroring the reference hcg API
    @property
    def data_parallel_group(self):
        return self.groups.get("dp")

    @property
    def sharding_parallel_group(self):
        return self.groups.get("sharding")

    @property
    def model_parallel_group(self):
        return self.groups.get("mp")

    @property
    def pipe_parallel_group(self):
        return self.groups.get("pp")

    @property
    def sep_parallel_group(self):
        return self.groups.get("sep")

    @property
    def dataset_world_size(self) -> int:
        """Ranks that see distinct data = dp * sharding (reference
        training_args.py dataset_world_size)."""
        return self.dp_degree * self.sharding_degree

    @property
    def dataset_rank(self) -> int:
        return self.coords.get("dp", 0) * self.sharding_degree + self.coords.get("sharding", 0)

    def is_last_pp_stage(self) -> bool:
        return self.coords.get("pp", 0) == self.pp_degree - 1

    def is_first_pp_stage(self) -> bool:
        return self.coords.get("pp", 0) == 0


def _unravel(rank: int, shape: List[int]) -> List[int]:
    coords = []
    for dim in reversed(shape):
        coords.append(rank % dim)
        rank //= dim
    return list(reversed(coords))


def _ravel(coords: List[int], shape: List[int]) -> int:
    rank = 0
    for c, dim in zip(coords, shape):
        rank = rank * dim + c
    return rank


def init_parallel_env(
    dp_degree: int = 1,
    pp_degree: int = 1,
    sharding_degree: int = 1,
    sep_degree: int = 1,
    mp_degree: int = 1,
    backend: Optional[str] = None,
    timeout_minutes: int = 30,
) -> Topology:
    """Initialize torch.distributed (if needed) and build the axis groups.

    Every rank must call this with identical degrees.  Degrees of 1 get no
    process group (group=None means "axis not parallel").
    """
    global _TOPOLOGY

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size > 1 and not dist.is_initialized():
        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        import datetime

        if backend == "nccl":
            local_rank = int(os.environ.get("LOCAL_RANK", "0"))
            torch.cuda.set_device(local_rank)
        dist.init_process_group(backend=backend, timeout=datetime.timedelta(minutes=timeout_minutes))

    rank = dist.get_rank() if dist.is_initialized() else 0
    world_size = dist.get_world_size() if dist.is_initialized() else 1

    degrees = {
        "dp": dp_degree,
        "pp": pp_degree,
        "sharding": sharding_degree,
        "sep": sep_degree,
        "mp": mp_degree,
    }
    prod = 1
    for d in degrees.values():
        prod *= d
    if prod != world_size:
        raise ValueError(
            f"Product of parallel degrees {degrees} = {prod} != world_size {world_size}"
        )

    shape = [degrees[a] for a in _AXES]
    my_coords = _unravel(rank, shape)

    topo = Topology(
        dp_degree=dp_degree,
        pp_degree=pp_degree,
        sharding_degree=sharding_degree,
        sep_degree=sep_degree,
        mp_degree=mp_degree,
        rank=rank,
        world_size=world_size,
    )
    topo.coords = {a: my_coords[i] for i, a in enumerate(_AXES)}

    # one group per axis: vary this axis, fix all others.  new_group must be
    # called by ALL ranks for EVERY group, in the same order.
    for ai, axis in enumerate(_AXES):
        if degrees[axis] == 1:
            topo.groups[axis] = None
            topo.group_ranks[axis] = [rank]
            continue
        my_group = None
        my_group_ranks = None
        # iterate all combinations of the other axes
        other_shape = shape[:ai] + shape[ai + 1:]
        total_other = 1
        for d in other_shape:
            total_other *= d
        for flat in range(total_other):
            other_coords = _unravel(flat, other_shape) if other_shape else []
            ranks = []
            for x in range(degrees[axis]):
                coords = other_coords[:ai] + [x] + other_coords[ai:]
                ranks.append(_ravel(coords, shape))
            group = dist.new_group(ranks=ranks) if dist.is_initialized() else None
            if rank in ranks:
                my_group = group
                my_group_ranks = ranks
        topo.groups[axis] = my_group
        topo.group_ranks[axis] = my_group_ranks or [rank]

    _TOPOLOGY = topo
    if rank == 0:
        logger.info(
            f"Initialized hybrid topology: world={world_size} "
            + " ".join(f"{a}={degrees[a]}" for a in _AXES)
        )
    return topo


def get_topology() -> Topology:
    global _TOPOLOGY
    if _TOPOLOGY is None:
        _TOPOLOGY = Topology()  # trivial single-process topology
    return _TOPOLOGY


def set_topology(topo: Topology) -> None:
    global _TOPOLOGY
    _TOPOLOGY = topo
