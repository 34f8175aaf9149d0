"""FusedAdamW: multi-tensor AdamW with fp32 master weights for bf16 params.

This is the MI355X-native equivalent of the reference's AMP-O2 optimizer
story (trainer.py:448 _wrap_amp_model: master weights + multi-tensor adamw).
On GPU the update runs through the gfx950 HIP multi-tensor kernel
(ops.fused_adamw); bf16 params keep an fp32 master copy that owns the
authoritative value.

Gradients arrive in the params' dtype (bf16); the kernel reads them as fp32.
"""
from __future__ import annotations


import torch
from torch.optim import Optimizer

from .. import ops


class FusedAdamW(Optimizer):
    def __init__(
        self,
        params,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        master_weights: bool = True,
        apply_decay_param_fun=None,
    ):
        defaults = dict(lr=lr, betas=betas, eps=eps, weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.master_weights = master_weights
        # reference: apply_decay_param_fun from trainer.py:1817 region —
        # layernorm/bias params get no weight decay
        self.apply_decay_param_fun = apply_decay_param_fun
        self._param_names = {}

    def _init_state(self, p):
        state = self.state[p]
        if "exp_avg" not in state:
            state["step"] = 0
            state["exp_avg"] = torch.zeros_like(p, dtype=torch.float32)
            state["exp_avg_sq"] = torch.zeros_like(p, dtype=torch.float32)
            if self.master_weights and p.dtype in (torch.bfloat16, torch.float16):
                state["master"] = p.detach().float().clone()
            else:
                state["master"] = None
        return state

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()

        for group in self.param_groups:
            beta1, beta2 = group["betas"]
            lr, eps, wd = group["lr"], group["eps"], group["weight_decay"]

            # bucket by (weight_decay applied?, step) for the multi-tensor kernel
            buckets = {}
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self._init_state(p)
                state["step"] += 1
                this_wd = wd
                name = getattr(p, "param_name", None)
                if self.apply_decay_param_fun is not None and name is not None:
                    if not self.apply_decay_param_fun(name):
                        this_wd = 0.0
                key = (this_wd, state["step"])
                buckets.setdefault(key, []).append((p, state))

            for (this_wd, step_t), items in buckets.items():
                params = [p for p, _ in items]
                grads = [p.grad for p, _ in items]
                exp_avgs = [s["exp_avg"] for _, s in items]
                exp_avg_sqs = [s["exp_avg_sq"] for _, s in items]
                masters = [s["master"] for _, s in items]
                if all(m is None for m in masters):
                    masters_arg = None
                else:
                    # mixed None is not allowed in the fused path; materialize
                    for i, m in enumerate(masters):
                        if m is None:
                            masters[i] = params[i].detach().float().clone()
                            items[i][1]["master"] = masters[i]
                    masters_arg = masters
                ops.fused_adamw(
                    params, grads, exp_avgs, exp_avg_sqs, masters_arg,
                    lr, beta1, beta2, eps, this_wd, step_t,
                )
        return loss

    def state_dict(self):
        sd = super().state_dict()
        return sd

    def zero_grad(self, set_to_none: bool = True):
        super().zero_grad(set_to_none=set_to_none)
