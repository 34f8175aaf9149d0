"""UnifiedTransformer (PLATO) family (reference:
paddlenlp/transformers/unified_transformer/modeling.py).

PREFIX-LM dialogue model: one pre-LN transformer stack where the dialogue
context attends BIDIRECTIONALLY and the response continues CAUSALLY — the
attention mask is the seq2seq (prefix) mask, built from ``prefix_len``.
Embeddings add word + position + token_type (+ optional role).  The tied
LM head generates the response; cached decode appends causal rows (a new
token sees the whole prefix and all generated tokens).
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ..configuration_utils import PretrainedConfig
from ..model_utils import PretrainedModel

__all__ = ["UnifiedTransformerConfig", "UnifiedTransformerModel",
           "UnifiedTransformerLMHeadModel"]


class UnifiedTransformerConfig(PretrainedConfig):
    model_type = "unified_transformer"

    def __init__(self, vocab_size=30004, hidden_size=768,
                 num_hidden_layers=12, num_attention_heads=12,
                 intermediate_size=3072, hidden_act="gelu",
                 hidden_dropout_prob=0.1, attention_probs_dropout_prob=0.1,
                 max_position_embeddings=512, type_vocab_size=2,
                 role_type_size=None, initializer_range=0.02,
                 layer_norm_eps=1e-12, pad_token_id=0, bos_token_id=1,
                 eos_token_id=2, mask_token_id=30000, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.intermediate_size = intermediate_size
        self.hidden_act = hidden_act
        self.hidden_dropout_prob = hidden_dropout_prob
        self.attention_probs_dropout_prob = attention_probs_dropout_prob
        self.max_position_embeddings = max_position_embeddings
        self.type_vocab_size = type_vocab_size
        self.role_type_size = role_type_size
        self.initializer_range = initializer_range
        self.layer_norm_eps = layer_norm_eps
        self.pad_token_id = pad_token_id
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.mask_token_id = mask_token_id


def prefix_lm_mask(S: int, prefix_len: torch.Tensor, past_len: int,
                   device) -> torch.Tensor:
    """Additive [B, 1, S, past+S] mask: position i attends to j iff
    j <= past_len + i (causal) OR j < prefix_len[b] (bidirectional prefix)."""
    total = past_len + S
    i = torch.arange(S, device=device)[:, None] + past_len
    j = torch.arange(total, device=device)[None, :]
    causal = j <= i                                      # [S, total]
    prefix = j[None] < prefix_len.to(device)[:, None, None]  # [B, S(total)]
    allowed = causal[None] | prefix
    return torch.where(allowed, 0.0, float("-inf"))[:, None]


class _PrefixAttention(nn.Module):
    def __init__(self, c: UnifiedTransformerConfig):
        super().__init__()
        h = c.hidden_size
        self.num_heads = c.num_attention_heads
        self.head_dim = h // self.num_heads
        self.qkv_proj = nn.Linear(h, 3 * h)
        self.out_proj = nn.Linear(h, h)
        self.dropout_p = c.attention_probs_dropout_prob

    def forward(self, x, mask, past_key_value=None, use_cache=False):
        B, S, H = x.shape
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q = q.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        k = k.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        v = v.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=2)
            v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None
        out = F.scaled_dot_product_attention(
            q, k, v, attn_mask=mask,
            dropout_p=self.dropout_p if self.training else 0.0)
        out = self.out_proj(out.transpose(1, 2).reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class _PrefixLayer(nn.Module):
    """Pre-LN block (the reference uses normalize_before=True)."""

    def __init__(self, c: UnifiedTransformerConfig):
        super().__init__()
        h = c.hidden_size
        self.attn_norm = nn.LayerNorm(h, eps=c.layer_norm_eps)
        self.attn = _PrefixAttention(c)
        self.mlp_norm = nn.LayerNorm(h, eps=c.layer_norm_eps)
        self.fc_in = nn.Linear(h, c.intermediate_size)
        self.fc_out = nn.Linear(c.intermediate_size, h)
        self.dropout = nn.Dropout(c.hidden_dropout_prob)

    def forward(self, x, mask, past_key_value=None, use_cache=False):
        a = self.attn(self.attn_norm(x), mask, past_key_value, use_cache)
        if use_cache:
            a, present = a
        x = x + self.dropout(a)
        x = x + self.dropout(self.fc_out(F.gelu(self.fc_in(self.mlp_norm(x)))))
        if use_cache:
            return x, present
        return x


class UnifiedTransformerPretrainedModel(PretrainedModel):
    config_class = UnifiedTransformerConfig
    base_model_prefix = "unified_transformer"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        elif isinstance(module, nn.LayerNorm):
            module.weight.data.fill_(1.0)
            module.bias.data.zero_()


class UnifiedTransformerModel(UnifiedTransformerPretrainedModel):
    def __init__(self, config: UnifiedTransformerConfig):
        super().__init__(config)
        h = config.hidden_size
        self.word_embeddings = nn.Embedding(config.vocab_size, h,
                                            padding_idx=config.pad_token_id)
        self.position_embeddings = nn.Embedding(config.max_position_embeddings, h)
        self.token_type_embeddings = nn.Embedding(config.type_vocab_size, h)
        self.role_embeddings = (nn.Embedding(config.role_type_size, h)
                                if config.role_type_size else None)
        self.embed_dropout = nn.Dropout(config.hidden_dropout_prob)
        self.layers = nn.ModuleList(
            [_PrefixLayer(config) for _ in range(config.num_hidden_layers)])
        self.norm = nn.LayerNorm(h, eps=config.layer_norm_eps)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids, token_type_ids=None, role_ids=None,
                prefix_len=None, past_key_values=None, use_cache=False):
        B, S = input_ids.shape
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None:
            past_len = past_key_values[0][0].shape[2]
        if prefix_len is None:
            # default: everything so far is bidirectional prefix (UniLM style
            # callers pass the real prompt length for training)
            prefix_len = torch.zeros(B, dtype=torch.long,
                                     device=input_ids.device)
        pos = torch.arange(past_len, past_len + S, device=input_ids.device)
        x = self.word_embeddings(input_ids) + self.position_embeddings(pos)
        if token_type_ids is not None:
            x = x + self.token_type_embeddings(token_type_ids)
        if role_ids is not None and self.role_embeddings is not None:
            x = x + self.role_embeddings(role_ids)
        x = self.embed_dropout(x)
        mask = prefix_lm_mask(S, prefix_len, past_len, input_ids.device)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, mask, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.norm(x)
        if use_cache:
            return x, presents
        return x


class UnifiedTransformerLMHeadModel(UnifiedTransformerPretrainedModel):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: UnifiedTransformerConfig):
        super().__init__(config)
        self.unified_transformer = UnifiedTransformerModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=False)
        self.lm_head.weight = self.unified_transformer.word_embeddings.weight
        self.init_weights()
        self.tie_weights()

    def tie_weights(self):
        self.lm_head.weight = self.unified_transformer.word_embeddings.weight

    def get_input_embeddings(self):
        return self.unified_transformer.word_embeddings

    def forward(self, input_ids, token_type_ids=None, role_ids=None,
                prefix_len=None, labels=None, past_key_values=None,
                use_cache=False):
        out = self.unified_transformer(input_ids, token_type_ids, role_ids,
                                       prefix_len, past_key_values, use_cache)
        if use_cache:
            hidden, presents = out
        else:
            hidden, presents = out, None
        logits = self.lm_head(hidden)
        if labels is not None:
            loss = ops.cross_entropy(
                logits.reshape(-1, logits.shape[-1]), labels.reshape(-1),
                -100, reduction="mean")
            return (loss, logits) if not use_cache else (loss, logits, presents)
        return logits if not use_cache else (logits, presents)

    @torch.no_grad()
    def generate(self, input_ids, token_type_ids=None, max_new_tokens=32,
                 do_sample=False, temperature=1.0, eos_token_id=None,
                 **kwargs):
        """Dialogue response generation: the whole input is the
        bidirectional prefix; new tokens extend causally."""
        eos = eos_token_id if eos_token_id is not None \
            else self.config.eos_token_id
        B, S = input_ids.shape
        device = input_ids.device
        prefix_len = torch.full((B,), S, dtype=torch.long, device=device)
        logits, past = self.forward(input_ids, token_type_ids,
                                    prefix_len=prefix_len, use_cache=True)
        cur = logits[:, -1].float()
        unfinished = torch.ones(B, dtype=torch.bool, device=device)
        tokens = []
        for _ in range(max_new_tokens):
            if do_sample:
                probs = (cur / max(temperature, 1e-6)).softmax(-1)
                token = torch.multinomial(probs, 1).squeeze(-1)
            else:
                token = cur.argmax(-1)
            token = torch.where(unfinished, token,
                                torch.full_like(token, self.config.pad_token_id))
            tokens.append(token)
            unfinished = unfinished & (token != eos)
            if not unfinished.any():
                break
            logits, past = self.forward(token[:, None], prefix_len=prefix_len,
                                        past_key_values=past, use_cache=True)
            cur = logits[:, -1].float()
        return torch.stack(tokens, dim=1), None
