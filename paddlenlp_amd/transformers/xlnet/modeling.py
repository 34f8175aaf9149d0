"""XLNet (reference: paddlenlp/transformers/xlnet/modeling.py).

Transformer-XL style encoder: relative positional attention with the
u/v bias split (r_w_bias / r_r_bias, modeling.py:65-74), learned segment
encoding (seg_embed, :66,:113), the rel_shift_bnij gather trick (:75-83),
and segment-level recurrence through cached `mems` (:1036 region).  The
permutation-LM two-stream machinery is pretrain-only; finetuning (the
reference's downstream heads) runs the content stream alone, which is
what this implementation provides.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["XLNetConfig", "XLNetModel", "XLNetLMHeadModel",
           "XLNetForSequenceClassification"]


class XLNetConfig(PretrainedConfig):
    model_type = "xlnet"

    attribute_map = {
        "d_model": "hidden_size", "n_layer": "num_hidden_layers",
        "n_head": "num_attention_heads", "d_inner": "intermediate_size",
        "num_classes": "num_labels",
    }

    def __init__(self, vocab_size=32000, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, ff_activation="gelu",
                 dropout=0.1, mem_len=0, clamp_len=-1,
                 initializer_range=0.02, layer_norm_eps=1e-12,
                 pad_token_id=5, num_labels=2, classifier_dropout=0.1,
                 **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.ff_activation = ff_activation
        self.dropout = dropout
        self.mem_len = mem_len
        self.clamp_len = clamp_len
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels
        self.classifier_dropout = classifier_dropout

    @property
    def d_head(self):
        return self.hidden_size // self.num_attention_heads


class XLNetRelativeAttention(nn.Module):
    """Content-stream relative attention (reference rel_attn_core :85-130)."""

    def __init__(self, config: XLNetConfig):
        super().__init__()
        h, nh, dh = config.hidden_size, config.num_attention_heads, config.d_head
        self.n_head, self.d_head = nh, dh
        self.scale = 1.0 / math.sqrt(dh)
        self.q = nn.Parameter(torch.empty(h, nh, dh))
        self.k = nn.Parameter(torch.empty(h, nh, dh))
        self.v = nn.Parameter(torch.empty(h, nh, dh))
        self.o = nn.Parameter(torch.empty(h, nh, dh))
        self.r = nn.Parameter(torch.empty(h, nh, dh))   # pos-emb projection
        self.r_w_bias = nn.Parameter(torch.zeros(nh, dh))  # u (content bias)
        self.r_r_bias = nn.Parameter(torch.zeros(nh, dh))  # v (position bias)
        self.r_s_bias = nn.Parameter(torch.zeros(nh, dh))
        self.seg_embed = nn.Parameter(torch.empty(2, nh, dh))
        self.layer_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.dropout)
        for p in (self.q, self.k, self.v, self.o, self.r, self.seg_embed):
            nn.init.normal_(p, std=config.initializer_range)

    @staticmethod
    def rel_shift_bnij(x, klen):
        # reference :75-83: pad-free reshape shift
        b, n, i, j = x.shape
        x = x.reshape(b, n, j, i)[:, :, 1:, :].reshape(b, n, i, j - 1)
        return x[:, :, :, :klen]

    def forward(self, h, r, seg_mat=None, attn_mask=None, mems=None):
        B, S, _ = h.shape
        cat = h if mems is None else torch.cat([mems, h], dim=1)
        klen = cat.shape[1]

        q_head = torch.einsum("bih,hnd->bind", h, self.q)
        k_head = torch.einsum("bih,hnd->bind", cat, self.k)
        v_head = torch.einsum("bih,hnd->bind", cat, self.v)
        r_head = torch.einsum("ih,hnd->ind", r, self.r)

        # ac: content score; bd: position score via rel-shift (:101-106)
        ac = torch.einsum("bind,bjnd->bnij", q_head + self.r_w_bias, k_head)
        bd = torch.einsum("bind,jnd->bnij", q_head + self.r_r_bias, r_head)
        bd = self.rel_shift_bnij(bd, klen)
        score = ac + bd
        if seg_mat is not None:
            # ef: same/diff-segment score (:113-115)
            ef = torch.einsum("bind,snd->bnis", q_head + self.r_s_bias,
                              self.seg_embed)
            ef = torch.einsum("bijs,bnis->bnij", seg_mat, ef)
            score = score + ef
        score = score * self.scale
        if attn_mask is not None:
            score = score + attn_mask
        probs = self.dropout(F.softmax(score, dim=-1))
        vec = torch.einsum("bnij,bjnd->bind", probs, v_head)
        out = torch.einsum("bind,hnd->bih", vec, self.o)
        return self.layer_norm(h + self.dropout(out))


class XLNetLayer(nn.Module):
    def __init__(self, config: XLNetConfig):
        super().__init__()
        self.rel_attn = XLNetRelativeAttention(config)
        h = config.hidden_size
        self.ff1 = nn.Linear(h, config.intermediate_size)
        self.ff2 = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.ff_activation]
        self.dropout = nn.Dropout(config.dropout)

    def forward(self, x, r, seg_mat=None, attn_mask=None, mems=None):
        x = self.rel_attn(x, r, seg_mat, attn_mask, mems)
        y = self.ff2(self.dropout(self.act(self.ff1(x))))
        return self.ff_norm(x + self.dropout(y))


class XLNetPretrainedModel(PretrainedModel):
    config_class = XLNetConfig
    base_model_prefix = "transformer"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class XLNetModel(XLNetPretrainedModel):
    def __init__(self, config: XLNetConfig):
        super().__init__(config)
        self.word_embedding = nn.Embedding(config.vocab_size,
                                           config.hidden_size)
        self.layers = nn.ModuleList(
            [XLNetLayer(config) for _ in range(config.num_hidden_layers)])
        self.dropout = nn.Dropout(config.dropout)

    def get_input_embeddings(self):
        return self.word_embedding

    def _pos_emb(self, qlen, klen, device, dtype):
        """Sinusoidal relative positions klen .. -qlen (reference
        relative_positional_encoding :1000 region)."""
        pos = torch.arange(klen, -qlen, -1.0, device=device)
        if self.config.clamp_len > 0:
            pos = pos.clamp(-self.config.clamp_len, self.config.clamp_len)
        h = self.config.hidden_size
        inv = 1.0 / (10000 ** (torch.arange(0, h, 2, device=device).float() / h))
        sin_in = torch.outer(pos, inv)
        return torch.cat([sin_in.sin(), sin_in.cos()], dim=-1).to(dtype)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                mems=None, use_mems: bool = False):
        B, S = input_ids.shape
        x = self.dropout(self.word_embedding(input_ids))
        mlen = mems[0].shape[1] if mems else 0
        klen = S + mlen
        r = self._pos_emb(S, klen, x.device, x.dtype)

        seg_mat = None
        if token_type_ids is not None:
            # same/diff one-hot; mems count as "diff" (reference :1090)
            mem_pad = torch.zeros(B, mlen, dtype=token_type_ids.dtype,
                                  device=x.device)
            cat_ids = torch.cat([mem_pad, token_type_ids], dim=1)
            diff = (token_type_ids[:, :, None] != cat_ids[:, None, :]).long()
            seg_mat = F.one_hot(diff, 2).to(x.dtype)

        mask = None
        if attention_mask is not None:
            pad = torch.ones(B, mlen, dtype=attention_mask.dtype,
                             device=x.device)
            m = torch.cat([pad, attention_mask], dim=1)
            mask = (1.0 - m[:, None, None, :].to(x.dtype)) * torch.finfo(x.dtype).min

        new_mems = []
        mem_len = self.config.mem_len
        for i, layer in enumerate(self.layers):
            if use_mems and mem_len > 0:
                cur = x if mems is None else torch.cat([mems[i], x], dim=1)
                new_mems.append(cur[:, -mem_len:].detach())
            x = layer(x, r, seg_mat, mask, mems[i] if mems else None)
        x = self.dropout(x)
        if use_mems:
            return x, new_mems
        return x, None


class XLNetLMHeadModel(XLNetPretrainedModel):
    _tied_weights_keys = ["lm_loss.weight"]

    def __init__(self, config: XLNetConfig):
        super().__init__(config)
        self.transformer = XLNetModel(config)
        self.lm_loss = nn.Linear(config.hidden_size, config.vocab_size)
        self.lm_loss.weight = self.transformer.word_embedding.weight

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                mems=None, use_mems=False, labels=None):
        seq, new_mems = self.transformer(input_ids, token_type_ids,
                                         attention_mask, mems, use_mems)
        logits = self.lm_loss(seq)
        if labels is not None:
            # labels pre-aligned by the caller (perm-LM targets / shifted ids)
            loss = F.cross_entropy(logits.view(-1, self.config.vocab_size),
                                   labels.view(-1), ignore_index=-100)
            return loss, logits
        return (logits, new_mems) if use_mems else logits


class XLNetForSequenceClassification(XLNetPretrainedModel):
    def __init__(self, config: XLNetConfig):
        super().__init__(config)
        self.transformer = XLNetModel(config)
        # reference sequence_summary: last-token summary + tanh projection
        self.sequence_summary = nn.Linear(config.hidden_size,
                                          config.hidden_size)
        self.dropout = nn.Dropout(config.classifier_dropout)
        self.logits_proj = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.transformer(input_ids, token_type_ids, attention_mask)
        if attention_mask is not None:
            last = attention_mask.long().sum(dim=1) - 1
        else:
            last = torch.full((input_ids.shape[0],), input_ids.shape[1] - 1,
                              device=input_ids.device, dtype=torch.long)
        pooled = seq[torch.arange(seq.shape[0], device=seq.device), last]
        logits = self.logits_proj(self.dropout(torch.tanh(
            self.sequence_summary(pooled))))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits
