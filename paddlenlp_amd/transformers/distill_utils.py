"""MiniLM-style distillation helpers.

Reference behavior: paddlenlp/transformers/distill_utils.py (to_distill,
calc_minilm_loss, calc_multi_relation_loss).  The reference monkey-patches
paddle.nn MultiHeadAttention forwards; here we use forward hooks on the
shared ``EncoderSelfAttention`` modules, capturing per-layer Q/K/V
([B, H, S, D]) into ``model.distill_qkv`` — no modeling-code changes.
"""
from __future__ import annotations

import math
from typing import List, Tuple

import torch
import torch.nn.functional as F

from .encoder import EncoderSelfAttention

__all__ = ["to_distill", "calc_minilm_loss", "calc_multi_relation_loss"]


def _capture_qkv(module: EncoderSelfAttention, args, kwargs, output):
    x = args[0] if args else kwargs["x"]
    B, S, H = x.shape
    q, k, v = module.qkv_proj(x).chunk(3, dim=-1)

    def heads(t):
        return t.view(B, S, module.num_heads, module.head_dim).transpose(1, 2)

    module._captured_qkv = (heads(q), heads(k), heads(v))


def to_distill(model, return_qkv: bool = True, layer_index: int = -1):
    """Instrument ``model`` so each forward records attention Q/K/V.

    After ``model(...)``, read ``model.distill_qkv`` -> list of (q, k, v)
    tuples (one per captured layer, [B, H, S, D]).  ``layer_index`` keeps
    only that encoder layer (-1 = last, the MiniLMv2 recipe); pass ``None``
    to keep all layers.
    """
    attns = [m for m in model.modules() if isinstance(m, EncoderSelfAttention)]
    if not attns:
        raise ValueError("no EncoderSelfAttention modules found to distill")
    if layer_index is not None:
        attns = [attns[layer_index]]
    handles = []
    for m in attns:
        handles.append(m.register_forward_hook(_capture_qkv, with_kwargs=True))
    model._distill_handles = handles
    model._distill_attns = attns

    cls = type(model)

    class _Distilled(cls):
        @property
        def distill_qkv(self) -> List[Tuple[torch.Tensor, ...]]:
            return [m._captured_qkv for m in self._distill_attns]

    _Distilled.__name__ = cls.__name__
    model.__class__ = _Distilled
    return model


def _relation_loss(loss_fct, s, t, num_relation_heads):
    """KL between self-relation matrices of student/teacher tensors.

    s, t: [B, H, S, D].  Optionally re-chunk heads to num_relation_heads
    (MiniLMv2: compare at a fixed relation-head count even when student and
    teacher widths differ)."""
    if num_relation_heads > 0 and num_relation_heads != s.shape[1]:
        B, H, S, D = s.shape
        s = s.transpose(1, 2).reshape(B, S, H * D) \
             .view(B, S, num_relation_heads, -1).transpose(1, 2)
    if num_relation_heads > 0 and num_relation_heads != t.shape[1]:
        B, H, S, D = t.shape
        t = t.transpose(1, 2).reshape(B, S, H * D) \
             .view(B, S, num_relation_heads, -1).transpose(1, 2)
    rel_s = s @ s.transpose(-1, -2) / math.sqrt(s.shape[-1])
    rel_t = t @ t.transpose(-1, -2) / math.sqrt(t.shape[-1])
    return loss_fct(F.log_softmax(rel_s, dim=-1), F.softmax(rel_t, dim=-1))


def calc_minilm_loss(loss_fct, s, t, attn_mask=None, num_relation_heads=0):
    """MiniLMv2 self-relation KL for one of Q/K/V (reference :119)."""
    return _relation_loss(loss_fct, s, t, num_relation_heads)


def calc_multi_relation_loss(loss_fct, s, t, attn_mask=None,
                             num_relation_heads=0, alpha=0.0, beta=0.0):
    """Weighted token-token + head-head + sample-sample relation loss
    (reference :31).  s, t: [B, H, S, D]."""
    loss_tt = _relation_loss(loss_fct, s, t, num_relation_heads) \
        if (1 - alpha - beta) > 0 else 0.0

    loss_hh = 0.0
    if alpha > 0:
        # relations across heads at each token: [B, S, H, D]
        sh, th = s.transpose(1, 2), t.transpose(1, 2)
        loss_hh = _relation_loss(loss_fct, sh, th, 0)

    loss_ss = 0.0
    if beta > 0:
        # relations across the batch per (head, position): [H, S, B, D]
        ss = s.permute(1, 2, 0, 3)
        ts = t.permute(1, 2, 0, 3)
        loss_ss = _relation_loss(loss_fct, ss, ts, 0)

    return (1 - alpha - beta) * loss_tt + alpha * loss_hh + beta * loss_ss
