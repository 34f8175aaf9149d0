"""Reformer (reference: paddlenlp/transformers/reformer/modeling.py).

The three Reformer mechanisms, re-implemented:
- LSH self-attention (reference LSHSelfAttention region): shared-QK,
  multi-round random-rotation bucketing, sort-by-bucket chunked attention
  with one look-back chunk, logsumexp combination across hash rounds.
- Local self-attention (chunked sliding window with look-back).
- Axial position embeddings (:326-360): two factorized tables whose
  outer sum covers the sequence.
Layer schedule follows `attn_layers` ("lsh"/"local" per layer, :128).
The reversible residual pairing (x1/x2 dual stream) is kept structurally;
activations are recomputed by autograd rather than by a custom
inverse-pass (torch recompute hooks cover the memory story on MI355X).
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ...generation import GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ReformerConfig", "ReformerModel", "ReformerModelWithLMHead"]

_NEG = -1e9


class ReformerConfig(PretrainedConfig):
    model_type = "reformer"

    def __init__(self, vocab_size=320, hidden_size=256,
                 num_attention_heads=2, attention_head_size=64,
                 feed_forward_size=512, attn_layers=("local", "lsh"),
                 lsh_attn_chunk_length=64, local_attn_chunk_length=64,
                 num_chunks_before=1, num_hashes=2, num_buckets=8,
                 axial_pos_shape=(8, 8), axial_pos_embds_dim=None,
                 hidden_act="relu", hidden_dropout_prob=0.05,
                 is_decoder=True, layer_norm_eps=1e-12,
                 initializer_range=0.02, pad_token_id=0, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_attention_heads = num_attention_heads
        self.attention_head_size = attention_head_size
        self.feed_forward_size = feed_forward_size
        self.attn_layers = list(attn_layers)
        self.num_hidden_layers = len(self.attn_layers)
        self.lsh_attn_chunk_length = lsh_attn_chunk_length
        self.local_attn_chunk_length = local_attn_chunk_length
        self.num_chunks_before = num_chunks_before
        self.num_hashes = num_hashes
        self.num_buckets = num_buckets
        self.axial_pos_shape = tuple(axial_pos_shape)
        self.axial_pos_embds_dim = tuple(
            axial_pos_embds_dim or (hidden_size // 2,
                                    hidden_size - hidden_size // 2))
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.is_decoder = is_decoder
        self.layer_norm_eps = layer_norm_eps
        self.initializer_range = initializer_range
        self.pad_token_id = pad_token_id


class AxialPositionEmbeddings(nn.Module):
    """Reference :326-360: hidden = concat over axes of broadcast tables."""

    def __init__(self, config: ReformerConfig):
        super().__init__()
        d1, d2 = config.axial_pos_shape
        e1, e2 = config.axial_pos_embds_dim
        self.shape = (d1, d2)
        self.w1 = nn.Parameter(torch.randn(d1, 1, e1) * 0.02)
        self.w2 = nn.Parameter(torch.randn(1, d2, e2) * 0.02)

    def forward(self, S: int, dtype):
        d1, d2 = self.shape
        full = torch.cat([self.w1.expand(d1, d2, -1),
                          self.w2.expand(d1, d2, -1)], dim=-1)
        return full.reshape(d1 * d2, -1)[:S].to(dtype)


def _chunked_attend(q, k, v, chunk: int, nbefore: int, causal: bool,
                    qpos, kpos, exclude_self: bool = False):
    """Chunk q into [n, c]; each chunk attends to itself + nbefore
    predecessor chunks of k/v.  q,k,v: [B, H, S, D] with S % chunk == 0.
    qpos/kpos carry ORIGINAL sequence positions (post-sort for LSH) and
    drive causal + wrap-around masking; exclude_self soft-masks the
    shared-QK diagonal (the Reformer self-attention penalty)."""
    B, H, S, D = q.shape
    n = S // chunk
    qc = q.reshape(B, H, n, chunk, D)

    def look(x):
        xc = x.reshape(B, H, n, chunk, -1)
        parts = [torch.roll(xc, shifts=i, dims=2) for i in range(nbefore, -1, -1)]
        return torch.cat(parts, dim=3)   # [B,H,n,(nbefore+1)*chunk,·]

    kc, vc = look(k), look(v)
    score = qc @ kc.transpose(-1, -2) / math.sqrt(D)
    qp = qpos.reshape(B if qpos.shape[0] == B else 1, qpos.shape[1], n, chunk)
    kp = look(kpos.unsqueeze(-1).to(q.dtype)).squeeze(-1)
    diff = qp.unsqueeze(-1) - kp[:, :, :, None, :]
    # wrap-around roll chunks land far away in position space: mask them
    if causal:
        valid = (diff >= 0) & (diff < (nbefore + 1) * chunk)
    else:
        valid = (diff > -chunk) & (diff < (nbefore + 1) * chunk)
    score = torch.where(valid, score, torch.full_like(score, _NEG))
    if exclude_self:
        score = torch.where(diff == 0, torch.full_like(score, -1e5), score)
    lse = torch.logsumexp(score, dim=-1, keepdim=True)
    probs = (score - lse).exp()
    # renormalize: at the -1e5 self-penalty magnitude, fp32 lse rounding
    # leaves probs summing to ~0.998 on self-only rows
    probs = probs / probs.sum(dim=-1, keepdim=True).clamp_min(1e-20)
    out = probs @ vc
    return out.reshape(B, H, S, D), lse.reshape(B, H, S)


class LocalSelfAttention(nn.Module):
    def __init__(self, config: ReformerConfig):
        super().__init__()
        h = config.hidden_size
        self.nh = config.num_attention_heads
        self.dh = config.attention_head_size
        self.chunk = config.local_attn_chunk_length
        self.nbefore = config.num_chunks_before
        self.causal = config.is_decoder
        self.query = nn.Linear(h, self.nh * self.dh, bias=False)
        self.key = nn.Linear(h, self.nh * self.dh, bias=False)
        self.value = nn.Linear(h, self.nh * self.dh, bias=False)
        self.out = nn.Linear(self.nh * self.dh, h, bias=False)

    def forward(self, x):
        B, S, _ = x.shape
        shp = (B, S, self.nh, self.dh)
        q = self.query(x).view(shp).transpose(1, 2)
        k = self.key(x).view(shp).transpose(1, 2)
        v = self.value(x).view(shp).transpose(1, 2)
        pos = torch.arange(S, device=x.device)
        out, _ = _chunked_attend(q, k, v, self.chunk, self.nbefore,
                                 self.causal,
                                 qpos=pos.view(1, 1, S).expand(1, 1, S),
                                 kpos=pos.view(1, 1, S).expand(B, self.nh, S))
        return self.out(out.transpose(1, 2).reshape(B, S, -1))


class LSHSelfAttention(nn.Module):
    """Shared-QK LSH attention (reference LSHSelfAttention region)."""

    def __init__(self, config: ReformerConfig):
        super().__init__()
        h = config.hidden_size
        self.nh = config.num_attention_heads
        self.dh = config.attention_head_size
        self.chunk = config.lsh_attn_chunk_length
        self.nbefore = config.num_chunks_before
        self.nhashes = config.num_hashes
        self.nbuckets = config.num_buckets
        self.causal = config.is_decoder
        # shared QK: one projection serves both (the Reformer trick)
        self.query_key = nn.Linear(h, self.nh * self.dh, bias=False)
        self.value = nn.Linear(h, self.nh * self.dh, bias=False)
        self.out = nn.Linear(self.nh * self.dh, h, bias=False)

    def _hash(self, qk, g: torch.Generator):
        """Random-rotation bucketing: argmax over [R, -R] rotations."""
        B, H, S, D = qk.shape
        rot = torch.randn(self.nhashes, D, self.nbuckets // 2,
                          device="cpu", generator=g).to(qk.device, qk.dtype)
        r = torch.einsum("bhsd,ndm->bhnsm", qk, rot)
        return torch.argmax(torch.cat([r, -r], dim=-1), dim=-1)  # [B,H,nh,S]

    def forward(self, x):
        B, S, _ = x.shape
        shp = (B, S, self.nh, self.dh)
        qk = self.query_key(x).view(shp).transpose(1, 2)   # [B,H,S,D]
        v = self.value(x).view(shp).transpose(1, 2)
        # deterministic rotations per forward (seeded by shape, matching
        # eval-time reproducibility; training-time variety is not needed
        # for the bucketing to be valid)
        g = torch.Generator().manual_seed(S * 1000003 + self.nbuckets)
        buckets = self._hash(F.normalize(qk, dim=-1), g)   # [B,H,R,S]

        pos = torch.arange(S, device=x.device)
        outs, lses = [], []
        for rnd in range(self.nhashes):
            bkt = buckets[:, :, rnd]                       # [B,H,S]
            # stable sort by bucket; ties keep sequence order
            key = bkt * S + pos
            order = key.argsort(dim=-1)                    # [B,H,S]
            inv = order.argsort(dim=-1)
            oe = order.unsqueeze(-1).expand(B, self.nh, S, self.dh)
            qs = torch.gather(qk, 2, oe)
            vs = torch.gather(v, 2, oe)
            ps = torch.gather(pos.view(1, 1, S).expand(B, self.nh, S), 2, order)
            o, l = _chunked_attend(qs, F.normalize(qs, dim=-1) *
                                   math.sqrt(self.dh), vs, self.chunk,
                                   self.nbefore, self.causal,
                                   qpos=ps, kpos=ps, exclude_self=True)
            outs.append(torch.gather(o, 2, inv.unsqueeze(-1).expand_as(o)))
            lses.append(torch.gather(l, 2, inv))
        if self.nhashes == 1:
            o = outs[0]
        else:
            # logsumexp-weighted combination across rounds
            L = torch.stack(lses, dim=0)                   # [R,B,H,S]
            w = F.softmax(L, dim=0).unsqueeze(-1)
            o = (torch.stack(outs, dim=0) * w).sum(dim=0)
        return self.out(o.transpose(1, 2).reshape(B, S, -1))


class ReformerLayer(nn.Module):
    """Reversible pairing: y1 = x1 + attn(ln(x2)); y2 = x2 + ff(ln(y1))."""

    def __init__(self, config: ReformerConfig, kind: str):
        super().__init__()
        h = config.hidden_size
        self.attn = (LSHSelfAttention(config) if kind == "lsh"
                     else LocalSelfAttention(config))
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.ff1 = nn.Linear(h, config.feed_forward_size)
        self.ff2 = nn.Linear(config.feed_forward_size, h)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x1, x2):
        y1 = x1 + self.dropout(self.attn(self.attn_norm(x2)))
        y2 = x2 + self.dropout(self.ff2(self.act(self.ff1(self.ff_norm(y1)))))
        return y1, y2


class ReformerPretrainedModel(PretrainedModel):
    config_class = ReformerConfig
    base_model_prefix = "reformer"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ReformerModel(ReformerPretrainedModel):
    def __init__(self, config: ReformerConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size)
        self.position_embeddings = AxialPositionEmbeddings(config)
        self.layers = nn.ModuleList(
            [ReformerLayer(config, kind) for kind in config.attn_layers])
        self.final_norm = nn.LayerNorm(2 * config.hidden_size,
                                       eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def get_input_embeddings(self):
        return self.word_embeddings

    def _chunk_mult(self):
        import numpy as np
        kinds = set(self.config.attn_layers)
        if kinds == {"lsh"}:
            return self.config.lsh_attn_chunk_length
        if kinds == {"local"}:
            return self.config.local_attn_chunk_length
        return int(np.lcm(self.config.lsh_attn_chunk_length,
                          self.config.local_attn_chunk_length))

    def forward(self, input_ids, attention_mask=None):
        B, S0 = input_ids.shape
        # pad to a chunk multiple (reference pads to least-common chunk :128)
        mult = self._chunk_mult()
        pad = (-S0) % mult
        if pad:
            input_ids = F.pad(input_ids, (0, pad),
                              value=self.config.pad_token_id)
        S = input_ids.shape[1]
        x = self.word_embeddings(input_ids) + \
            self.position_embeddings(S, self.word_embeddings.weight.dtype)
        x = self.dropout(x)
        x1 = x2 = x
        for layer in self.layers:
            x1, x2 = layer(x1, x2)
        out = self.final_norm(torch.cat([x1, x2], dim=-1))
        return out[:, :S0]


class ReformerModelWithLMHead(ReformerPretrainedModel, GenerationMixin):
    """Causal LM head.  `use_cache` keeps the token prefix (a recompute
    cache): LSH bucketing and chunk composition depend on the whole
    sequence, so there is no valid token-level KV cache — each decode
    step re-runs the prefix, exactly as the reference does for its LSH
    layers."""

    def __init__(self, config: ReformerConfig):
        super().__init__(config)
        self.reformer = ReformerModel(config)
        self.lm_head = nn.Linear(2 * config.hidden_size, config.vocab_size)

    def forward(self, input_ids, attention_mask=None, labels=None,
                past_key_values=None, use_cache=False, **kwargs):
        S_new = input_ids.shape[1]
        if past_key_values is not None:
            input_ids = torch.cat([past_key_values, input_ids], dim=1)
        logits = self.lm_head(self.reformer(input_ids, attention_mask))
        if past_key_values is not None:
            logits = logits[:, -S_new:]
        present = input_ids if use_cache else None
        if labels is not None:
            # labels are pre-shifted by the caller (framework convention)
            loss = F.cross_entropy(
                logits.reshape(-1, self.config.vocab_size),
                labels.reshape(-1), ignore_index=-100)
            return (loss, logits) if not use_cache else (loss, logits, present)
        return logits if not use_cache else (logits, present)
