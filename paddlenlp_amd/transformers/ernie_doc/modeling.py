"""ERNIE-Doc (reference: paddlenlp/transformers/ernie_doc/modeling.py).

Long-document encoder with SEGMENT RECURRENCE: each layer keeps a
`memory_len` cache of its input states; attention runs over
[memory | current] with Transformer-XL relative-position scores plus a
relative TASK (segment-role) embedding (reference attention :139-168,
ErnieDocEncoder memory update :247-277).  The retrospective feed (same
doc twice: skim pass fills memories, retrospective pass re-reads) is a
calling convention on top of `forward(..., memories=)`.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import ACT2FN, init_encoder_weights
from ..model_utils import PretrainedModel

__all__ = ["ErnieDocConfig", "ErnieDocModel",
           "ErnieDocForSequenceClassification"]


class ErnieDocConfig(PretrainedConfig):
    model_type = "ernie_doc"

    def __init__(self, vocab_size=50265, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 memory_len=128, task_type_vocab_size=3,
                 initializer_range=0.02, layer_norm_eps=1e-12,
                 pad_token_id=1, num_labels=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.memory_len = memory_len
        self.task_type_vocab_size = task_type_vocab_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


class ErnieDocAttention(nn.Module):
    """Rel-pos + rel-task attention over [memory | current]."""

    def __init__(self, config: ErnieDocConfig):
        super().__init__()
        h, nh, dh = (config.hidden_size, config.num_attention_heads,
                     config.head_dim)
        self.nh, self.dh = nh, dh
        self.q_proj = nn.Linear(h, h)
        self.k_proj = nn.Linear(h, h)
        self.v_proj = nn.Linear(h, h)
        self.r_proj = nn.Linear(h, h)      # relative position projection
        self.t_proj = nn.Linear(h, h)      # relative task projection
        self.out_proj = nn.Linear(h, h)
        self.r_w_bias = nn.Parameter(torch.zeros(nh, dh))
        self.r_r_bias = nn.Parameter(torch.zeros(nh, dh))
        self.r_t_bias = nn.Parameter(torch.zeros(nh, dh))

    @staticmethod
    def _rel_shift(x, klen):
        b, n, q, l = x.shape
        x = x.reshape(b, n, l, q)[:, :, 1:, :].reshape(b, n, q, l - 1)
        return x[:, :, :, :klen]

    def forward(self, x, rel_pos, rel_task, memory=None):
        B, S, H = x.shape
        cat = x if memory is None else torch.cat([memory, x], dim=1)
        klen = cat.shape[1]
        q = self.q_proj(x).view(B, S, self.nh, self.dh)
        k = self.k_proj(cat).view(B, klen, self.nh, self.dh)
        v = self.v_proj(cat).view(B, klen, self.nh, self.dh)
        r = self.r_proj(rel_pos).view(-1, self.nh, self.dh)
        t = self.t_proj(rel_task).view(-1, self.nh, self.dh)

        ac = torch.einsum("bind,bjnd->bnij", q + self.r_w_bias, k)
        bd = self._rel_shift(
            torch.einsum("bind,jnd->bnij", q + self.r_r_bias, r), klen)
        # task scores: rel_task table is tiny (same/other segment roles);
        # broadcast over keys via their task ids handled by the caller
        # through rel_task ordering [klen, H]
        ef = torch.einsum("bind,jnd->bnij", q + self.r_t_bias,
                          t[:klen])
        score = (ac + bd + ef) / math.sqrt(self.dh)
        probs = score.softmax(-1)
        out = torch.einsum("bnij,bjnd->bind", probs, v)
        return self.out_proj(out.reshape(B, S, H))


class ErnieDocLayer(nn.Module):
    def __init__(self, config: ErnieDocConfig):
        super().__init__()
        h = config.hidden_size
        self.attn = ErnieDocAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.ff_in = nn.Linear(h, config.intermediate_size)
        self.ff_out = nn.Linear(config.intermediate_size, h)
        self.ff_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]

    def forward(self, x, rel_pos, rel_task, memory=None):
        x = self.attn_norm(x + self.attn(x, rel_pos, rel_task, memory))
        return self.ff_norm(x + self.ff_out(self.act(self.ff_in(x))))


class ErnieDocPretrainedModel(PretrainedModel):
    config_class = ErnieDocConfig
    base_model_prefix = "ernie_doc"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ErnieDocModel(ErnieDocPretrainedModel):
    def __init__(self, config: ErnieDocConfig):
        super().__init__(config)
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h,
                                            padding_idx=config.pad_token_id)
        # table long enough for memory + 2*segment relative range (:313)
        self.pos_embeddings = nn.Embedding(
            config.max_position_embeddings * 2 + config.memory_len, h)
        self.task_embeddings = nn.Embedding(config.task_type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [ErnieDocLayer(config)
             for _ in range(config.num_hidden_layers)])
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, memories=None):
        """memories: per-layer [B, memory_len, H] from the previous
        segment (or None).  Returns (sequence_output, new_memories)."""
        B, S = input_ids.shape
        mlen = memories[0].shape[1] if memories else 0
        klen = S + mlen
        x = self.embed_norm(self.word_embeddings(input_ids))
        # relative positions klen-1 .. -(S-1), clipped into the table
        rel = torch.arange(klen + S - 1, -1, -1, device=input_ids.device)
        rel_pos = self.pos_embeddings(
            rel.clamp(max=self.pos_embeddings.num_embeddings - 1))
        rel_task = self.task_embeddings(
            torch.zeros(klen + S, dtype=torch.long,
                        device=input_ids.device))
        new_memories = []
        mem_len = self.config.memory_len
        for i, layer in enumerate(self.layers):
            mem = memories[i] if memories else None
            cur = x if mem is None else torch.cat([mem, x], dim=1)
            new_memories.append(cur[:, -mem_len:].detach())
            x = layer(x, rel_pos, rel_task, mem)
        return x, new_memories


class ErnieDocForSequenceClassification(ErnieDocPretrainedModel):
    def __init__(self, config: ErnieDocConfig):
        super().__init__(config)
        self.ernie_doc = ErnieDocModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)
        self.init_weights()

    def forward(self, input_ids, memories=None, labels=None):
        seq, new_mem = self.ernie_doc(input_ids, memories)
        # ERNIE-Doc pools the LAST token (its [CLS] sits at the end)
        logits = self.classifier(self.dropout(seq[:, -1]))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits, new_mem
