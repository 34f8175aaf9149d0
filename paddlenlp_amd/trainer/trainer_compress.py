"""Model compression: DynaBERT width pruning + distillation, PTQ with algo
grid search, QAT fake-quant training, embedding quantization.

Reference behavior: paddlenlp/trainer/trainer_compress.py (1k LoC over
paddleslim OFA/PTQ/QAT).  MI355X design: the same four strategies as plain
torch transformations over the shared encoder core — head/FFN importance is
measured with activation*gradient saliency, PTQ calibrates static activation
scales with forward hooks (abs_max / avg / mse grid), QAT uses
straight-through fake-quant, and "embeddings" quantizes embedding tables to
int8 rows.  No paddleslim: everything here is self-contained.
"""
from __future__ import annotations

import copy
import os
from typing import Callable, Dict, List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..transformers.encoder import EncoderLayer, EncoderSelfAttention
from ..utils.log import logger
from .compression_args import CompressionArguments


# ---------------------------------------------------------------------------
# PTQ: static activation-scale calibration + simulated a8w8 linears
# ---------------------------------------------------------------------------
class _ActObserver:
    """Collects an activation scale for one linear input.

    Scalar algos: abs_max / avg / mse.  Vector algos (reference
    llm/experimental/observer/{channel_wise,abs_max_headwise}.py):
    channel_wise keeps one scale per input channel; abs_max_headwise
    keeps one per attention head (head count via `heads`)."""

    def __init__(self, algo: str, heads: int = 0):
        self.algo = algo
        self.heads = heads
        self.max_seen = 0.0
        self.maxes: List[float] = []
        self.samples: List[torch.Tensor] = []
        self.vec_max: Optional[torch.Tensor] = None

    def update(self, x: torch.Tensor):
        if self.algo in ("channel_wise", "abs_max_headwise"):
            v = x.detach().abs().reshape(-1, x.shape[-1]).amax(dim=0).float()
            if self.algo == "abs_max_headwise":
                v = v.reshape(self.heads, -1).amax(dim=-1)
            self.vec_max = v if self.vec_max is None else                 torch.maximum(self.vec_max, v.to(self.vec_max.device))
            return
        m = float(x.detach().abs().amax())
        self.max_seen = max(self.max_seen, m)
        self.maxes.append(m)
        if self.algo == "mse" and len(self.samples) < 8:
            self.samples.append(x.detach().flatten()[:4096].float().cpu())

    def scale(self):
        if self.algo in ("channel_wise", "abs_max_headwise"):
            v = self.vec_max if self.vec_max is not None else torch.ones(1)
            return v.clamp(min=1e-8)
        if self.algo == "abs_max":
            s = self.max_seen
        elif self.algo == "avg":
            s = sum(self.maxes) / max(1, len(self.maxes))
        elif self.algo == "mse":
            x = torch.cat(self.samples) if self.samples else torch.zeros(1)
            best, s = float("inf"), self.max_seen
            for frac in torch.linspace(0.5, 1.0, 11):
                cand = self.max_seen * float(frac)
                if cand <= 0:
                    continue
                step = cand / 127.0
                q = torch.clamp(torch.round(x / step), -127, 127) * step
                err = float(((x - q) ** 2).mean())
                if err < best:
                    best, s = err, cand
        else:
            raise ValueError(f"unknown PTQ algo {self.algo!r}")
        return max(s, 1e-8)


class A8W8Linear(nn.Module):
    """Simulated static-scale int8xint8 linear: activations quantized with the
    calibrated scale, weights per-channel int8 (reference PTQ output form)."""

    def __init__(self, linear: nn.Linear, act_scale: float):
        super().__init__()
        self.in_features = linear.in_features
        self.out_features = linear.out_features
        w = linear.weight.data
        w_scale = w.abs().amax(dim=1).clamp(min=1e-8).float() / 127.0
        self.register_buffer(
            "quant_weight",
            torch.clamp(torch.round(w.float() / w_scale[:, None]), -127, 127).to(torch.int8))
        self.register_buffer("weight_scale", w_scale)
        self.register_buffer("act_scale", torch.tensor(float(act_scale)))
        self.bias = (nn.Parameter(linear.bias.data.clone())
                     if linear.bias is not None else None)

    def forward(self, x):
        step = self.act_scale / 127.0
        xq = torch.clamp(torch.round(x.float() / step), -127, 127)
        y = (xq @ self.quant_weight.float().t()) * (step * self.weight_scale)
        if self.bias is not None:
            y = y + self.bias.float()
        return y.to(x.dtype)


def post_training_quantization(model: nn.Module, calib_dataloader, algo: str = "avg",
                               batch_nums: int = 4,
                               skip: tuple = ("lm_head", "classifier", "pooler")):
    """Calibrate activation scales over `batch_nums` batches, then swap every
    non-skipped nn.Linear for a simulated a8w8 linear.  Returns the model."""
    observers: Dict[str, _ActObserver] = {}
    hooks = []
    for name, mod in model.named_modules():
        if isinstance(mod, nn.Linear) and not any(s in name for s in skip):
            obs = observers[name] = _ActObserver(algo)
            hooks.append(mod.register_forward_pre_hook(
                lambda m, args, _obs=obs: _obs.update(args[0])))
    model.eval()
    with torch.no_grad():
        for i, batch in enumerate(calib_dataloader):
            if i >= batch_nums:
                break
            inputs = {k: v for k, v in batch.items() if k != "labels"}
            model(**inputs)
    for h in hooks:
        h.remove()
    for name, obs in observers.items():
        parent_name, _, leaf = name.rpartition(".")
        parent = model.get_submodule(parent_name) if parent_name else model
        setattr(parent, leaf, A8W8Linear(getattr(parent, leaf), obs.scale()))
    logger.info(f"PTQ({algo}): quantized {len(observers)} linears")
    return model


# ---------------------------------------------------------------------------
# QAT: straight-through fake quantization
# ---------------------------------------------------------------------------
def fake_quant(x: torch.Tensor, num_bits: int = 8,
               per_channel: bool = False) -> torch.Tensor:
    qmax = 2 ** (num_bits - 1) - 1
    if per_channel:
        scale = x.detach().abs().amax(dim=tuple(range(1, x.dim())),
                                      keepdim=True).clamp(min=1e-8) / qmax
    else:
        scale = x.detach().abs().amax().clamp(min=1e-8) / qmax
    q = torch.clamp(torch.round(x / scale), -qmax, qmax) * scale
    return x + (q - x).detach()  # straight-through estimator


class QATLinear(nn.Module):
    """nn.Linear with fake-quantized weight (per-channel) and activation
    (per-tensor); gradients flow via STE."""

    def __init__(self, linear: nn.Linear):
        super().__init__()
        self.in_features = linear.in_features
        self.out_features = linear.out_features
        self.weight = linear.weight
        self.bias = linear.bias

    def forward(self, x):
        return F.linear(fake_quant(x), fake_quant(self.weight, per_channel=True),
                        self.bias)


def quant_aware_training(model: nn.Module,
                         skip: tuple = ("lm_head", "classifier", "pooler")):
    """Swap linears for QAT fake-quant linears (training continues outside)."""
    n = 0
    for name, mod in list(model.named_modules()):
        if isinstance(mod, nn.Linear) and not any(s in name for s in skip):
            parent_name, _, leaf = name.rpartition(".")
            parent = model.get_submodule(parent_name) if parent_name else model
            setattr(parent, leaf, QATLinear(mod))
            n += 1
    logger.info(f"QAT: wrapped {n} linears with fake-quant")
    return model


# ---------------------------------------------------------------------------
# embeddings: int8 row-quantized embedding tables
# ---------------------------------------------------------------------------
class QuantEmbedding(nn.Module):
    def __init__(self, emb: nn.Embedding):
        super().__init__()
        w = emb.weight.data
        scale = w.abs().amax(dim=1).clamp(min=1e-8).float() / 127.0
        self.register_buffer(
            "quant_weight",
            torch.clamp(torch.round(w.float() / scale[:, None]), -127, 127).to(torch.int8))
        self.register_buffer("scale", scale)
        self.num_embeddings, self.embedding_dim = w.shape
        self.padding_idx = emb.padding_idx
        self.out_dtype = w.dtype

    def forward(self, ids):
        return (self.quant_weight[ids].float()
                * self.scale[ids].unsqueeze(-1)).to(self.out_dtype)


def quantize_embeddings(model: nn.Module):
    n = 0
    for name, mod in list(model.named_modules()):
        if isinstance(mod, nn.Embedding):
            parent_name, _, leaf = name.rpartition(".")
            parent = model.get_submodule(parent_name) if parent_name else model
            setattr(parent, leaf, QuantEmbedding(mod))
            n += 1
    logger.info(f"quantized {n} embedding tables to int8 rows")
    return model


# ---------------------------------------------------------------------------
# DynaBERT: activation*gradient head / FFN-channel saliency -> width pruning
# ---------------------------------------------------------------------------
def _collect_saliency(model: nn.Module, dataloader, num_batches: int = 4):
    """Returns {layer_name: (head_saliency [H], ffn_saliency [I])} for every
    EncoderLayer, using |activation * grad| summed over calibration batches."""
    layers = {n: m for n, m in model.named_modules() if isinstance(m, EncoderLayer)}
    sal: Dict[str, list] = {n: [None, None] for n in layers}
    hooks = []

    def attn_hook(layer_name, attn: EncoderSelfAttention):
        # out_proj's input is the per-head attention context: hook its grad
        def pre_hook(mod, args):
            x = args[0]
            if not x.requires_grad:
                return
            H, D = attn.num_heads, attn.head_dim
            def grad_hook(g, _x=x):
                s = (_x.detach() * g.detach()).abs()
                s = s.reshape(*s.shape[:-1], H, D).sum(dim=(0, 1, 3))
                prev = sal[layer_name][0]
                sal[layer_name][0] = s if prev is None else prev + s
            x.register_hook(grad_hook)
        return pre_hook

    def ffn_hook(layer_name):
        def pre_hook(mod, args):
            x = args[0]  # input of fc_out: [B, S, I]
            if not x.requires_grad:
                return
            def grad_hook(g, _x=x):
                s = (_x.detach() * g.detach()).abs().sum(dim=(0, 1))
                prev = sal[layer_name][1]
                sal[layer_name][1] = s if prev is None else prev + s
            x.register_hook(grad_hook)
        return pre_hook

    for name, layer in layers.items():
        hooks.append(layer.self_attn.out_proj.register_forward_pre_hook(
            attn_hook(name, layer.self_attn)))
        hooks.append(layer.fc_out.register_forward_pre_hook(ffn_hook(name)))

    model.train()
    for i, batch in enumerate(dataloader):
        if i >= num_batches:
            break
        out = model(**batch)
        loss = out[0] if isinstance(out, tuple) else out
        model.zero_grad(set_to_none=True)
        loss.backward()
    model.zero_grad(set_to_none=True)
    for h in hooks:
        h.remove()
    return {n: (v[0], v[1]) for n, v in sal.items()}


def _slice_linear(linear: nn.Linear, out_idx=None, in_idx=None) -> nn.Linear:
    w = linear.weight.data
    b = linear.bias.data if linear.bias is not None else None
    if out_idx is not None:
        w = w[out_idx]
        if b is not None:
            b = b[out_idx]
    if in_idx is not None:
        w = w[:, in_idx]
    new = nn.Linear(w.shape[1], w.shape[0], bias=b is not None)
    new.weight.data.copy_(w)
    if b is not None:
        new.bias.data.copy_(b)
    return new


def dynabert_prune(model: nn.Module, dataloader, width_mult: float = 0.75,
                   num_batches: int = 4) -> nn.Module:
    """Width-prune every EncoderLayer to `width_mult` of its attention heads
    and FFN channels, keeping the most salient (reference _dynabert ladder,
    compressed to the prune step; the distillation finetune runs outside)."""
    model = copy.deepcopy(model)
    saliency = _collect_saliency(model, dataloader, num_batches)
    for name, mod in model.named_modules():
        if not isinstance(mod, EncoderLayer):
            continue
        head_sal, ffn_sal = saliency[name]
        attn = mod.self_attn
        H, D = attn.num_heads, attn.head_dim
        keep_h = max(1, int(round(H * width_mult)))
        keep_i = max(1, int(round(ffn_sal.numel() * width_mult)))
        top_h = torch.sort(torch.topk(head_sal, keep_h).indices).values
        top_i = torch.sort(torch.topk(ffn_sal, keep_i).indices).values

        # qkv_proj rows: same head selection in each of the q/k/v sections
        hidden = attn.out_proj.in_features
        head_rows = (top_h[:, None] * D
                     + torch.arange(D, device=top_h.device)).flatten()
        qkv_rows = torch.cat([head_rows, head_rows + hidden, head_rows + 2 * hidden])
        attn.qkv_proj = _slice_linear(attn.qkv_proj, out_idx=qkv_rows)
        attn.out_proj = _slice_linear(attn.out_proj, in_idx=head_rows)
        attn.num_heads = keep_h

        mod.fc_in = _slice_linear(mod.fc_in, out_idx=top_i)
        mod.fc_out = _slice_linear(mod.fc_out, in_idx=top_i)
    logger.info(f"dynabert: pruned to width_mult={width_mult}")
    return model


def distill_step(student: nn.Module, teacher: nn.Module, batch: dict,
                 temperature: float = 2.0) -> torch.Tensor:
    """One knowledge-distillation loss: KL(student || teacher) on logits
    (+ hard-label CE when labels are present)."""
    inputs = {k: v for k, v in batch.items() if k != "labels"}
    with torch.no_grad():
        t_out = teacher(**inputs)
        t_logits = t_out[0] if isinstance(t_out, tuple) else t_out
    s_out = student(**inputs)
    s_logits = s_out[0] if isinstance(s_out, tuple) else s_out
    T = temperature
    kd = F.kl_div(F.log_softmax(s_logits / T, dim=-1),
                  F.softmax(t_logits / T, dim=-1),
                  reduction="batchmean") * T * T
    if "labels" in batch:
        out = student(**batch)
        kd = kd + (out[0] if isinstance(out, tuple) else out)
    return kd


# ---------------------------------------------------------------------------
# the `compress` entry point (reference trainer_compress.compress:51)
# ---------------------------------------------------------------------------
def compress(trainer, args: Optional[CompressionArguments] = None,
             custom_evaluate: Optional[Callable] = None):
    """Run the configured compression strategies against trainer.model using
    trainer's train dataloader for calibration/distillation.  Saves each
    stage under args.output_dir and returns {strategy: model}."""
    args = args or CompressionArguments()
    results = {}
    dataloader = trainer.get_train_dataloader()
    model = trainer.model
    os.makedirs(args.output_dir, exist_ok=True)

    for strategy in args.strategies:
        if strategy == "dynabert":
            teacher = model
            for width in args.width_mult_list:
                student = dynabert_prune(teacher, dataloader, width)
                opt = torch.optim.AdamW(student.parameters(), lr=args.learning_rate)
                for i, batch in enumerate(dataloader):
                    if i >= args.logging_steps:
                        break
                    loss = distill_step(student, teacher, batch)
                    opt.zero_grad(set_to_none=True)
                    loss.backward()
                    opt.step()
                results[f"dynabert_{width}"] = student
        elif strategy == "ptq":
            best, best_metric = None, None
            for algo in args.algo_list:
                for bn in args.batch_num_list:
                    q = post_training_quantization(
                        copy.deepcopy(model), dataloader, algo, bn)
                    metric = (custom_evaluate(q, dataloader)
                              if custom_evaluate else -_calib_loss(q, dataloader))
                    logger.info(f"PTQ grid algo={algo} batches={bn}: {metric:.4f}")
                    if best_metric is None or metric > best_metric:
                        best, best_metric = q, metric
            results["ptq"] = best
        elif strategy == "qat":
            results["qat"] = quant_aware_training(copy.deepcopy(model))
        elif strategy == "embeddings":
            results["embeddings"] = quantize_embeddings(copy.deepcopy(model))
    return results


def _calib_loss(model, dataloader, num_batches: int = 2) -> float:
    model.eval()
    total, n = 0.0, 0
    with torch.no_grad():
        for i, batch in enumerate(dataloader):
            if i >= num_batches:
                break
            out = model(**batch)
            loss = out[0] if isinstance(out, tuple) else out
            total += float(loss)
            n += 1
    return total / max(1, n)
