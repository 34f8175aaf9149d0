"""ZeRO stage 3: parameter sharding with gather-on-demand.

Reference behavior: paddle group_sharded_parallel FULL_SHARD (SURVEY §2.2,
trainer.py:2047).  Each rank stores a flat 1/N shard of every parameter;
module pre-forward hooks all-gather the full weight just-in-time, the weight
is re-sharded after forward, re-gathered before that module's backward, and
the produced full gradient is immediately reduce-scattered into a sharded
gradient buffer.  The optimizer then steps on 1-D shards (so moments and
master weights are sharded too), and the next forward gathers the updated
shards — no post-step broadcast.

Mechanics note: the gather/reshard swaps `p.data` storage.  Autograd's saved
reference is the Parameter itself and `.data` assignment does not bump the
version counter, so restoring the full storage in the backward pre-hook
makes the saved tensor valid again (the classic ZeRO-3 data-swap design).

xGMI note: gathers/reduce-scatters are per-parameter collectives on the
sharding group; module-level prefetch batching is the planned overlap
refinement.
"""
from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

from ..utils.log import logger


class Zero3Engine:
    def __init__(self, model: nn.Module, group, grad_dtype: Optional[torch.dtype] = None):
        self.model = model
        self.group = group
        self.world = dist.get_world_size(group) if group is not None else 1
        self.rank = dist.get_rank(group) if group is not None else 0
        self.params: List[nn.Parameter] = []
        self._hooks = []

        seen = set()
        for p in model.parameters():
            if not p.requires_grad or id(p) in seen:
                continue
            seen.add(id(p))
            self._shard_param(p, grad_dtype)
            self.params.append(p)
        self._install_hooks(model)
        total = sum(p._z3_numel for p in self.params)
        local = sum(p._z3_shard.numel() for p in self.params)
        logger.info(f"ZeRO-3: rank {self.rank}/{self.world} stores "
                    f"{local:,} of {total:,} param elements")

    # ------------------------------------------------------------------
    def _shard_param(self, p: nn.Parameter, grad_dtype):
        full = p.data.detach().contiguous()
        numel = full.numel()
        pad = (self.world - numel % self.world) % self.world
        flat = torch.empty(numel + pad, dtype=full.dtype, device=full.device)
        flat[:numel] = full.reshape(-1)
        if pad:
            flat[numel:] = 0
        shard_n = flat.numel() // self.world
        shard = flat[self.rank * shard_n:(self.rank + 1) * shard_n].clone()
        p._z3_shape = full.shape
        p._z3_numel = numel
        p._z3_shard = shard
        p._z3_grad_shard = torch.zeros_like(shard, dtype=grad_dtype or full.dtype)
        p._z3_full = None
        p.data = shard  # 1-D shard view between gathers (optimizer steps on this)

    def _gather(self, p: nn.Parameter):
        if p._z3_full is not None:
            return
        shard = p._z3_shard
        if self.world == 1:
            full_flat = shard
        else:
            full_flat = torch.empty(shard.numel() * self.world,
                                    dtype=shard.dtype, device=shard.device)
            dist.all_gather_into_tensor(full_flat, shard.contiguous(), group=self.group)
        p._z3_full = full_flat
        p.data = full_flat[:p._z3_numel].view(p._z3_shape)

    def _reshard(self, p: nn.Parameter):
        if p._z3_full is None:
            return
        p.data = p._z3_shard
        p._z3_full = None

    # ------------------------------------------------------------------
    def _install_hooks(self, model: nn.Module):
        # module hooks go on EVERY module with direct trainable params (tied
        # params appear in several modules: gather is idempotent and each
        # consumer must be able to re-gather after an earlier reshard)
        for module in model.modules():
            direct = [p for p in module.parameters(recurse=False) if p.requires_grad]
            if not direct:
                continue

            def pre_fwd(mod, args, ps=tuple(direct)):
                for p in ps:
                    self._gather(p)

            def post_fwd(mod, args, out, ps=tuple(direct)):
                # reshard immediately; the backward pre-hook re-gathers
                for p in ps:
                    self._reshard(p)
                return out

            def pre_bwd(mod, grad_output, ps=tuple(direct)):
                for p in ps:
                    self._gather(p)

            self._hooks.append(module.register_forward_pre_hook(pre_fwd))
            self._hooks.append(module.register_forward_hook(post_fwd))
            self._hooks.append(module.register_full_backward_pre_hook(pre_bwd))

        # per-parameter: reduce-scatter the full grad as soon as it is ready
        for p in self.params:
            self._hooks.append(
                p.register_post_accumulate_grad_hook(self._make_grad_hook(p)))

    def _make_grad_hook(self, p):
        def hook(param):
            grad = param.grad
            if grad is None:
                return
            flat = grad.reshape(-1)
            pad = p._z3_shard.numel() * self.world - flat.numel()
            if pad:
                flat = torch.cat([flat, flat.new_zeros(pad)])
            if self.world == 1:
                rs = flat.clone()
            else:
                rs = torch.empty_like(p._z3_shard)
                dist.reduce_scatter_tensor(rs, flat.contiguous(), group=self.group)
                rs.div_(self.world)
            p._z3_grad_shard.add_(rs.to(p._z3_grad_shard.dtype))
            param.grad = None
            self._reshard(p)
        return hook

    # ------------------------------------------------------------------
    # trainer integration (mirrors ZeroShardedEngine's call points)
    # ------------------------------------------------------------------
    @property
    def stage(self):
        return 3

    def reduce_gradients_and_step_pre(self):
        """Grads were reduced eagerly; expose the sharded grads to the optimizer."""
        for p in self.params:
            p.grad = p._z3_grad_shard.to(p._z3_shard.dtype) \
                if p._z3_grad_shard.dtype != p._z3_shard.dtype else p._z3_grad_shard

    def step_post(self):
        for p in self.params:
            p.grad = None
            p._z3_grad_shard.zero_()

    def grad_norm_sq(self) -> torch.Tensor:
        """Global grad norm^2 (sharded sum + all-reduce)."""
        device = self.params[0]._z3_shard.device
        total = torch.zeros((), dtype=torch.float32, device=device)
        for p in self.params:
            total += p._z3_grad_shard.float().pow(2).sum()
        if self.group is not None:
            dist.all_reduce(total, group=self.group)
        return total

    def clip_grads(self, max_norm: float):
        norm = self.grad_norm_sq().sqrt()
        clip = (max_norm / (norm + 1e-6)).clamp(max=1.0)
        for p in self.params:
            p._z3_grad_shard.mul_(clip.to(p._z3_grad_shard.dtype))

    @torch.no_grad()
    def gather_full_state_dict(self) -> Dict[str, torch.Tensor]:
        """Materialize the full (unsharded) state dict for checkpointing."""
        for p in self.params:
            self._gather(p)
        sd = {k: v.detach().cpu().clone() for k, v in self.model.state_dict().items()}
        for p in self.params:
            self._reshard(p)
        return sd
