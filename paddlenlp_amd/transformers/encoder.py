"""Shared bidirectional transformer-encoder building blocks.

One MI355X-native encoder core reused by the BERT / ERNIE / RoBERTa /
ELECTRA families (reference: paddlenlp/transformers/{bert,ernie,roberta,
electra}/modeling.py each re-implement this stack; here it is a single
module and the families configure it).

Post-LN encoder (residual -> add -> LayerNorm), full bidirectional
attention through the paddlenlp_amd.ops flash-attention seam
([B, S, H, D] layout, causal=False).  A padding ``attention_mask`` of
shape [B, S] (1 = keep) routes through the masked fallback path.
"""
from __future__ import annotations

from typing import Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops

ACT2FN = {
    "gelu": lambda x: F.gelu(x),
    "gelu_tanh": lambda x: F.gelu(x, approximate="tanh"),
    "gelu_new": lambda x: F.gelu(x, approximate="tanh"),
    "relu": F.relu,
    "silu": F.silu,
    "tanh": torch.tanh,
}


def expand_padding_mask(attention_mask: torch.Tensor, dtype: torch.dtype) -> torch.Tensor:
    """[B, S] 1/0 keep-mask -> additive [B, 1, 1, S] mask in `dtype`."""
    mask = attention_mask[:, None, None, :].to(dtype)
    return (1.0 - mask) * torch.finfo(dtype).min


class EncoderEmbeddings(nn.Module):
    """word + learned-position (+ optional token-type / task-type) embeddings
    with LayerNorm + dropout."""

    def __init__(self, config, position_offset: int = 0):
        super().__init__()
        embed_dim = getattr(config, "embedding_size", None) or config.hidden_size
        self.word_embeddings = nn.Embedding(
            config.vocab_size, embed_dim, padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, embed_dim)
        self.position_offset = position_offset
        type_vocab = getattr(config, "type_vocab_size", 0)
        self.token_type_embeddings = (
            nn.Embedding(type_vocab, embed_dim) if type_vocab else None)
        task_vocab = getattr(config, "task_type_vocab_size", 0)
        use_task_id = getattr(config, "use_task_id", False)
        self.task_type_embeddings = (
            nn.Embedding(task_vocab, embed_dim) if (task_vocab and use_task_id) else None)
        self.layer_norm = nn.LayerNorm(embed_dim, eps=config.layer_norm_eps)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, input_ids, token_type_ids=None, position_ids=None,
                task_type_ids=None):
        B, S = input_ids.shape
        if position_ids is None:
            position_ids = torch.arange(
                self.position_offset, self.position_offset + S,
                device=input_ids.device).unsqueeze(0)
        x = self.word_embeddings(input_ids) + self.position_embeddings(position_ids)
        if self.token_type_embeddings is not None:
            if token_type_ids is None:
                token_type_ids = torch.zeros_like(input_ids)
            x = x + self.token_type_embeddings(token_type_ids)
        if self.task_type_embeddings is not None:
            if task_type_ids is None:
                task_type_ids = torch.zeros_like(input_ids)
            x = x + self.task_type_embeddings(task_type_ids)
        return self.dropout(self.layer_norm(x))


class EncoderSelfAttention(nn.Module):
    """Fused-QKV bidirectional self-attention ([B, S, H, D] layout)."""

    def __init__(self, config):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = h // self.num_heads
        self.qkv_proj = nn.Linear(h, 3 * h, bias=True)
        self.out_proj = nn.Linear(h, h, bias=True)
        self.dropout_p = config.attention_probs_dropout_prob

    def forward(self, x, attention_mask: Optional[torch.Tensor] = None):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim)
        k = k.view(B, S, self.num_heads, self.head_dim)
        v = v.view(B, S, self.num_heads, self.head_dim)
        if attention_mask is None:
            out = ops.flash_attention(q, k, v, causal=False)
        else:
            add_mask = expand_padding_mask(attention_mask, q.dtype)
            out = F.scaled_dot_product_attention(
                q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
                attn_mask=add_mask,
                dropout_p=self.dropout_p if self.training else 0.0,
            ).transpose(1, 2)
        return self.out_proj(out.reshape(B, S, self.num_heads * self.head_dim))


class EncoderLayer(nn.Module):
    """Post-LN: x = LN(x + attn(x)); x = LN(x + mlp(x))."""

    def __init__(self, config):
        super().__init__()
        h = config.hidden_size
        self.self_attn = EncoderSelfAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size, bias=True)
        self.fc_out = nn.Linear(config.intermediate_size, h, bias=True)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, attention_mask=None):
        x = self.attn_norm(x + self.dropout(self.self_attn(x, attention_mask)))
        mlp = self.fc_out(self.act(self.fc_in(x)))
        return self.mlp_norm(x + self.dropout(mlp))


class TransformerEncoder(nn.Module):
    def __init__(self, config):
        super().__init__()
        self.layers = nn.ModuleList(
            [EncoderLayer(config) for _ in range(config.num_hidden_layers)])

    def forward(self, x, attention_mask=None, output_hidden_states=False):
        all_hidden = [x] if output_hidden_states else None
        for layer in self.layers:
            x = layer(x, attention_mask)
            if output_hidden_states:
                all_hidden.append(x)
        if output_hidden_states:
            return x, all_hidden
        return x


class EncoderPooler(nn.Module):
    """tanh projection of the [CLS] hidden state."""

    def __init__(self, config):
        super().__init__()
        self.dense = nn.Linear(config.hidden_size, config.hidden_size)

    def forward(self, hidden_states):
        return torch.tanh(self.dense(hidden_states[:, 0]))


class LMPredictionHead(nn.Module):
    """transform (dense + act + LN) -> decoder tied to word embeddings."""

    def __init__(self, config, embedding_weights: Optional[torch.Tensor] = None):
        super().__init__()
        embed_dim = getattr(config, "embedding_size", None) or config.hidden_size
        self.dense = nn.Linear(config.hidden_size, embed_dim)
        self.act = ACT2FN[config.hidden_act]
        self.layer_norm = nn.LayerNorm(embed_dim, eps=config.layer_norm_eps)
        self.decoder = nn.Linear(embed_dim, config.vocab_size, bias=True)
        if embedding_weights is not None:
            self.decoder.weight = embedding_weights

    def forward(self, hidden_states):
        x = self.layer_norm(self.act(self.dense(hidden_states)))
        return self.decoder(x)


def init_encoder_weights(module, std: float):
    if isinstance(module, nn.Linear):
        module.weight.data.normal_(mean=0.0, std=std)
        if module.bias is not None:
            module.bias.data.zero_()
    elif isinstance(module, nn.Embedding):
        module.weight.data.normal_(mean=0.0, std=std)
        if module.padding_idx is not None:
            module.weight.data[module.padding_idx].zero_()
    elif isinstance(module, nn.LayerNorm):
        module.weight.data.fill_(1.0)
        module.bias.data.zero_()
