"""LLM quantization flows: shift, smooth-quant / AWQ scale search, autoclip,
PTQ, GPTQ.

Reference behavior: llm/utils/quant.py (apply_shift :119, apply_smooth :139,
apply_autoclip :187, apply_ptq :392, apply_gptq :467 over paddleslim).
MI355X design: self-contained torch implementations — activation statistics
come from forward hooks over a calibration dataloader; smoothing folds into
the RMSNorm scale feeding each norm->linear pair (the only place per-channel
division is free); GPTQ runs the OBQ column sweep on a damped Hessian from
calibration activations.
"""
from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import torch
import torch.nn as nn

from paddlenlp_amd.quantization import QuantizationLinear
from paddlenlp_amd.utils.log import logger


# ---------------------------------------------------------------------------
# calibration statistics
# ---------------------------------------------------------------------------
def collect_activation_stats(model: nn.Module, dataloader, num_batches: int = 8,
                             names: Optional[List[str]] = None):
    """Per-linear input stats: {name: {"absmax": [in], "mean": [in]}}."""
    stats: Dict[str, dict] = {}
    hooks = []
    for name, mod in model.named_modules():
        if not isinstance(mod, nn.Linear):
            continue
        if names is not None and name not in names:
            continue
        entry = stats[name] = {"absmax": None, "mean": None, "count": 0}

        def pre_hook(m, args, _e=entry):
            x = args[0].detach()
            flat = x.reshape(-1, x.shape[-1]).float()
            amax = flat.abs().amax(0)
            mean = flat.mean(0)
            n = flat.shape[0]
            if _e["absmax"] is None:
                _e["absmax"], _e["mean"], _e["count"] = amax, mean, n
            else:
                _e["absmax"] = torch.maximum(_e["absmax"], amax)
                tot = _e["count"] + n
                _e["mean"] = (_e["mean"] * _e["count"] + mean * n) / tot
                _e["count"] = tot

        hooks.append(mod.register_forward_pre_hook(pre_hook))
    model.eval()
    with torch.no_grad():
        for i, batch in enumerate(dataloader):
            if i >= num_batches:
                break
            inputs = {k: v for k, v in batch.items() if k != "labels"}
            model(**inputs)
    for h in hooks:
        h.remove()
    return stats


def _norm_linear_pairs(model: nn.Module) -> List[Tuple[nn.Module, List[Tuple[str, nn.Linear]]]]:
    """(norm_module, [(name, linear), ...]) for each norm whose output feeds
    the linears directly (llama-style decoder layers: input_layernorm ->
    qkv/q/k/v, post_attention_layernorm -> gate_up/gate/up)."""
    pairs = []
    for lname, layer in model.named_modules():
        attn = getattr(layer, "self_attn", None)
        mlp = getattr(layer, "mlp", None)
        in_ln = getattr(layer, "input_layernorm", None)
        post_ln = getattr(layer, "post_attention_layernorm", None)
        if attn is not None and in_ln is not None and hasattr(in_ln, "weight"):
            lins = [(f"{lname}.self_attn.{n}", getattr(attn, n))
                    for n in ("qkv_proj", "q_proj", "k_proj", "v_proj")
                    if isinstance(getattr(attn, n, None), nn.Linear)]
            if lins:
                pairs.append((in_ln, lins))
        if mlp is not None and post_ln is not None and hasattr(post_ln, "weight"):
            lins = [(f"{lname}.mlp.{n}", getattr(mlp, n))
                    for n in ("gate_up_fused_proj", "gate_proj", "up_proj")
                    if isinstance(getattr(mlp, n, None), nn.Linear)]
            if lins:
                pairs.append((post_ln, lins))
    return pairs


# ---------------------------------------------------------------------------
# shift / smooth (outlier suppression + SmoothQuant/AWQ)
# ---------------------------------------------------------------------------
@torch.no_grad()
def apply_shift(model: nn.Module, dataloader, num_batches: int = 8):
    """Outlier-suppression shift: subtract the per-channel activation mean
    ahead of each norm->linear pair and compensate in the linear bias
    (reference apply_shift :119).  Requires the norm to expose a bias or the
    linears to have biases; linears without bias get one."""
    from paddlenlp_amd.transformers.llama.modeling import LlamaRMSNorm

    pairs = _norm_linear_pairs(model)
    # only norms whose forward APPLIES shift_bias may carry the shift —
    # an inert buffer would silently corrupt outputs (bias compensation
    # without the matching subtraction)
    pairs = [(n, l) for n, l in pairs if isinstance(n, LlamaRMSNorm)]
    names = [n for _, lins in pairs for n, _ in lins]
    stats = collect_activation_stats(model, dataloader, num_batches, names)
    shifted = 0
    for norm, lins in pairs:
        first = next(n for n, _ in lins if n in stats)
        z = stats[first]["mean"].to(norm.weight.dtype)  # [in]
        for name, lin in lins:
            if lin.bias is None:
                lin.bias = nn.Parameter(
                    torch.zeros(lin.out_features, dtype=lin.weight.dtype,
                                device=lin.weight.device))
            # y = W(x - z) + (b + W z): fold the mean into the bias
            lin.bias.add_((lin.weight.float() @ z.float()).to(lin.bias.dtype))
        if not hasattr(norm, "shift_bias"):
            norm.register_buffer("shift_bias", -z.clone())
        else:
            norm.shift_bias.add_(-z)
        shifted += 1
    logger.info(f"shift: processed {shifted} norm->linear groups")
    return model


@torch.no_grad()
def apply_smooth(model: nn.Module, dataloader, alpha: float = 0.5,
                 num_batches: int = 8, do_awq: bool = False,
                 awq_grid: int = 5):
    """SmoothQuant scale migration (reference apply_smooth :139):
    s_j = absmax(X_j)^alpha / absmax(W_:,j)^(1-alpha); activations divide by
    s (folded into the preceding RMSNorm weight), weights multiply by s.
    With do_awq=True, alpha is grid-searched per pair to minimize int8
    weight-quant output error (AWQ-style)."""
    from paddlenlp_amd.transformers.llama.modeling import LlamaRMSNorm

    pairs = _norm_linear_pairs(model)
    # scale folds into norms whose output is LINEAR in the weight
    # (out = xhat * w); (1+w)-style norms (Gemma) cannot absorb a division
    pairs = [(n, l) for n, l in pairs if isinstance(n, LlamaRMSNorm)]
    names = [n for _, lins in pairs for n, _ in lins]
    stats = collect_activation_stats(model, dataloader, num_batches, names)
    for norm, lins in pairs:
        first = next(n for n, _ in lins if n in stats)
        a_max = stats[first]["absmax"].clamp(min=1e-5)          # [in]
        w_max = torch.cat([l.weight for _, l in lins]).abs().amax(0).clamp(min=1e-5)

        def scale_for(al):
            return (a_max ** al) / (w_max ** (1 - al))

        if do_awq:
            best_s, best_err = None, float("inf")
            for al in torch.linspace(0.2, 0.8, awq_grid):
                s = scale_for(float(al)).clamp(min=1e-5)
                err = 0.0
                for _, lin in lins:
                    w = lin.weight.float() * s[None, :]
                    q = torch.clamp(torch.round(
                        w / (w.abs().amax(1, keepdim=True) / 127)), -127, 127)
                    wq = q * (w.abs().amax(1, keepdim=True) / 127)
                    # error in the UNscaled domain, weighted by act magnitude
                    err += float((((wq - w) / s[None, :]) ** 2 * a_max[None, :] ** 2).sum())
                if err < best_err:
                    best_err, best_s = err, s
            s = best_s
        else:
            s = scale_for(alpha).clamp(min=1e-5)
        norm.weight.div_(s.to(norm.weight.dtype))
        for _, lin in lins:
            lin.weight.mul_(s[None, :].to(lin.weight.dtype))
    logger.info(f"smooth({'awq' if do_awq else f'alpha={alpha}'}): "
                f"{len(pairs)} norm->linear groups")
    return model


@torch.no_grad()
def apply_autoclip(model: nn.Module, dataloader, num_batches: int = 4,
                   n_grid: int = 10, max_shrink: float = 0.5):
    """AWQ AutoClip (reference apply_autoclip :187): per-output-channel
    search of the weight clip threshold minimizing int8 output MSE against
    calibration activations."""
    stats = collect_activation_stats(model, dataloader, num_batches)
    clipped = 0
    for name, mod in model.named_modules():
        if not isinstance(mod, nn.Linear) or name not in stats:
            continue
        x_amax = stats[name]["absmax"]                   # [in]
        w = mod.weight.float()                           # [out, in]
        orig_max = w.abs().amax(1, keepdim=True).clamp(min=1e-8)
        best = orig_max.clone()
        # proxy output: sum_j |w_ij| * absmax_j (cheap rank-1 calibration)
        ref = (w * x_amax[None, :]).sum(1)
        best_err = torch.full_like(ref, float("inf"))
        for gi in range(n_grid):
            shrink = 1.0 - max_shrink * gi / max(1, n_grid - 1)
            cmax = orig_max * shrink
            wc = w.clamp(-cmax, cmax)
            scale = cmax / 127.0
            wq = torch.clamp(torch.round(wc / scale), -127, 127) * scale
            err = ((wq - w) * x_amax[None, :]).pow(2).sum(1)
            better = err < best_err
            best_err = torch.where(better, err, best_err)
            best = torch.where(better[:, None], cmax, best)
        mod.weight.copy_(w.clamp(-best, best).to(mod.weight.dtype))
        clipped += 1
    logger.info(f"autoclip: clipped {clipped} linears")
    return model


# ---------------------------------------------------------------------------
# PTQ / GPTQ
# ---------------------------------------------------------------------------
def apply_ptq(model: nn.Module, dataloader, algo: str = "avg",
              num_batches: int = 4):
    """Static a8w8 PTQ (reference apply_ptq :392) — calibrated activation
    scales + per-channel int8 weights via trainer_compress."""
    from paddlenlp_amd.trainer.trainer_compress import post_training_quantization

    return post_training_quantization(model, dataloader, algo, num_batches)


@torch.no_grad()
def gptq_quantize_weight(w: torch.Tensor, H: torch.Tensor, bits: int = 4,
                         percdamp: float = 0.01,
                         group_size: int = -1) -> torch.Tensor:
    """OBQ/GPTQ column sweep: quantize w [out, in] column-by-column against
    the damped Hessian H = X^T X [in, in], propagating the compensation
    W[:, j:] -= err * Hinv[j, j:] / Hinv[j, j].  Returns the dequantized
    weight (same dtype/shape)."""
    out_f, in_f = w.shape
    W = w.float().clone()
    H = H.float().clone()
    dead = torch.diag(H) == 0
    H[dead, dead] = 1.0
    W[:, dead] = 0
    damp = percdamp * torch.diag(H).mean()
    H += torch.eye(in_f, device=H.device) * damp
    # Cholesky-based inverse (upper) as in GPTQ
    Hinv = torch.linalg.cholesky(
        torch.cholesky_inverse(torch.linalg.cholesky(H)), upper=True)

    qmax = 2 ** (bits - 1) - 1
    gsize = in_f if group_size in (-1, 0) else group_size
    scale = None
    for j in range(in_f):
        if j % gsize == 0:
            blk = W[:, j:j + gsize]
            scale = blk.abs().amax(1, keepdim=True).clamp(min=1e-8) / qmax
        col = W[:, j]
        q = torch.clamp(torch.round(col / scale[:, 0]), -qmax, qmax) * scale[:, 0]
        err = (col - q) / Hinv[j, j]
        W[:, j] = q
        if j + 1 < in_f:
            W[:, j + 1:] -= err[:, None] * Hinv[j, j + 1:][None, :]
    return W.to(w.dtype)


@torch.no_grad()
def apply_gptq(model: nn.Module, dataloader, bits: int = 4,
               num_batches: int = 8, group_size: int = -1,
               skip: tuple = ("lm_head",)):
    """GPTQ over every linear (reference apply_gptq :467): calibration
    Hessians from input activations, column-sweep quantization in place."""
    hessians: Dict[str, torch.Tensor] = {}
    hooks = []
    for name, mod in model.named_modules():
        if not isinstance(mod, nn.Linear) or any(s in name for s in skip):
            continue

        def pre_hook(m, args, _n=name):
            x = args[0].detach().reshape(-1, args[0].shape[-1]).float()
            h = x.t() @ x
            if _n in hessians:
                hessians[_n] += h
            else:
                hessians[_n] = h

        hooks.append(mod.register_forward_pre_hook(pre_hook))
    model.eval()
    with torch.no_grad():
        for i, batch in enumerate(dataloader):
            if i >= num_batches:
                break
            model(**{k: v for k, v in batch.items() if k != "labels"})
    for h in hooks:
        h.remove()

    n = 0
    for name, mod in model.named_modules():
        if name in hessians:
            mod.weight.copy_(gptq_quantize_weight(
                mod.weight, hessians[name], bits, group_size=group_size))
            n += 1
    logger.info(f"gptq: quantized {n} linears to int{bits}")
    return model


def quantize_to_weight_only(model: nn.Module, algo: str = "weight_only_int8",
                            skip: tuple = ("lm_head",)):
    """Swap linears for packed QuantizationLinear after the flows above."""
    n = 0
    for name, mod in list(model.named_modules()):
        if not isinstance(mod, nn.Linear) or any(s in name for s in skip):
            continue
        parent_name, _, leaf = name.rpartition(".")
        parent = model.get_submodule(parent_name) if parent_name else model
        setattr(parent, leaf, QuantizationLinear.from_linear(mod, algo))
        n += 1
    logger.info(f"packed {n} linears as {algo}")
    return model
