"""ConvBERT (reference: paddlenlp/transformers/convbert/modeling.py).

Mixed-attention encoder: `head_ratio` of the attention heads are replaced
by span-based dynamic convolution — a depthwise-separable conv over the
keys produces per-position conv kernels (softmax over kernel taps) that
filter an unfolded value window.  Self-attention runs on the remaining
heads; both halves concatenate back to the hidden size.
"""
from __future__ import annotations


import torch
import torch.nn as nn
import torch.nn.functional as F

from ..configuration_utils import PretrainedConfig
from ..encoder import (
    ACT2FN,
    EncoderPooler,
    LMPredictionHead,
    expand_padding_mask,
    init_encoder_weights,
)
from ..model_utils import PretrainedModel

__all__ = ["ConvBertConfig", "ConvBertModel",
           "ConvBertForSequenceClassification", "ConvBertForMaskedLM"]


class ConvBertConfig(PretrainedConfig):
    model_type = "convbert"

    def __init__(self, vocab_size=30522, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, max_position_embeddings=512,
                 type_vocab_size=2, head_ratio=2, conv_kernel_size=9,
                 initializer_range=0.02, layer_norm_eps=1e-12,
                 pad_token_id=0, num_labels=2, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.head_ratio = head_ratio
        self.conv_kernel_size = conv_kernel_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.num_labels = num_labels


class ConvBertMixedAttention(nn.Module):
    def __init__(self, config: ConvBertConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads // config.head_ratio
        self.head_dim = h // config.num_attention_heads
        self.all_head = self.num_heads * self.head_dim
        self.ks = config.conv_kernel_size
        self.query = nn.Linear(h, self.all_head)
        self.key = nn.Linear(h, self.all_head)
        self.value = nn.Linear(h, self.all_head)
        # conv branch: separable conv over keys -> dynamic kernels
        self.key_conv_attn = nn.Conv1d(h, self.all_head, self.ks,
                                       padding=self.ks // 2, groups=h // 8
                                       if h % 8 == 0 else 1)
        self.conv_kernel_layer = nn.Linear(self.all_head,
                                           self.num_heads * self.ks)
        self.conv_out_layer = nn.Linear(h, self.all_head)
        self.dense = nn.Linear(2 * self.all_head, h)

    def forward(self, x, attention_mask=None):
        B, S, H = x.shape
        q = self.query(x)
        k = self.key(x)
        v = self.value(x)

        # ---- self-attention half ----
        shp = (B, S, self.num_heads, self.head_dim)
        qh = q.view(shp).transpose(1, 2)
        kh = k.view(shp).transpose(1, 2)
        vh = v.view(shp).transpose(1, 2)
        add_mask = (expand_padding_mask(attention_mask, x.dtype)
                    if attention_mask is not None else None)
        attn = F.scaled_dot_product_attention(qh, kh, vh, attn_mask=add_mask)
        attn = attn.transpose(1, 2).reshape(B, S, self.all_head)

        # ---- span dynamic conv half ----
        mixed = self.key_conv_attn(x.transpose(1, 2)).transpose(1, 2)
        kernel = self.conv_kernel_layer(mixed * q)          # [B,S,heads*ks]
        kernel = F.softmax(kernel.view(B, S, self.num_heads, self.ks), dim=-1)
        conv_v = self.conv_out_layer(x)                     # [B,S,all_head]
        # unfold a ks-window of values at each position
        pad = self.ks // 2
        win = F.pad(conv_v, (0, 0, pad, pad))
        win = win.unfold(1, self.ks, 1)                     # [B,S,all_head,ks]
        win = win.reshape(B, S, self.num_heads, self.head_dim, self.ks)
        conv = torch.einsum("bshdk,bshk->bshd", win, kernel)
        conv = conv.reshape(B, S, self.all_head)

        return self.dense(torch.cat([attn, conv], dim=-1))


class ConvBertLayer(nn.Module):
    def __init__(self, config: ConvBertConfig):
        super().__init__()
        h = config.hidden_size
        self.attn = ConvBertMixedAttention(config)
        self.attn_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.fc_in = nn.Linear(h, config.intermediate_size)
        self.fc_out = nn.Linear(config.intermediate_size, h)
        self.mlp_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.act = ACT2FN[config.hidden_act]
        self.dropout = nn.Dropout(config.hidden_dropout_prob)

    def forward(self, x, attention_mask=None):
        x = self.attn_norm(x + self.dropout(self.attn(x, attention_mask)))
        y = self.fc_out(self.act(self.fc_in(x)))
        return self.mlp_norm(x + self.dropout(y))


class ConvBertPretrainedModel(PretrainedModel):
    config_class = ConvBertConfig
    base_model_prefix = "convbert"

    def _init_weights(self, module):
        init_encoder_weights(module, self.config.initializer_range)


class ConvBertModel(ConvBertPretrainedModel):
    def __init__(self, config: ConvBertConfig):
        super().__init__(config)
        h = config.hidden_size
        self.embeddings = nn.Embedding(config.vocab_size, h,
                                       padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(
            config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.embed_norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.layers = nn.ModuleList(
            [ConvBertLayer(config) for _ in range(config.num_hidden_layers)])
        self.pooler = EncoderPooler(config)

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, token_type_ids=None, attention_mask=None):
        S = input_ids.shape[1]
        pos = torch.arange(S, device=input_ids.device)
        x = self.embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        x = self.embed_norm(x)
        for layer in self.layers:
            x = layer(x, attention_mask)
        return x, self.pooler(x)


class ConvBertForSequenceClassification(ConvBertPretrainedModel):
    def __init__(self, config: ConvBertConfig):
        super().__init__(config)
        self.convbert = ConvBertModel(config)
        self.dropout = nn.Dropout(config.hidden_dropout_prob)
        self.classifier = nn.Linear(config.hidden_size, config.num_labels)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        _, pooled = self.convbert(input_ids, token_type_ids, attention_mask)
        logits = self.classifier(self.dropout(pooled))
        if labels is not None:
            return F.cross_entropy(logits, labels.view(-1)), logits
        return logits


class ConvBertForMaskedLM(ConvBertPretrainedModel):
    _tied_weights_keys = ["cls.decoder.weight"]

    def __init__(self, config: ConvBertConfig):
        super().__init__(config)
        self.convbert = ConvBertModel(config)
        self.cls = LMPredictionHead(config, self.convbert.embeddings.weight)

    def forward(self, input_ids, token_type_ids=None, attention_mask=None,
                labels=None):
        seq, _ = self.convbert(input_ids, token_type_ids, attention_mask)
        logits = self.cls(seq)
        if labels is not None:
            loss = F.cross_entropy(
                logits.view(-1, self.config.vocab_size), labels.view(-1),
                ignore_index=-100)
            return loss, logits
        return logits
