"""ChatGLM (v1, GLM-6B) family (reference:
paddlenlp/transformers/chatglm/modeling.py).

GLM prefix decoder with three signature quirks:
- alpha-scaled post-LN residuals: y = alpha * LN_input + sublayer(LN_input)
  with alpha = sqrt(2 * num_layers) (reference chatglm:385-392);
- 2-D rotary: each head's dims are split in half — the first half rotates
  with the POSITION id (context index, frozen at mask_pos for generated
  tokens), the second half with the BLOCK id (0 in context, 1.. for the
  generation) (reference chatglm:207-225);
- prefix-LM attention: the prompt is bidirectional, generation is causal
  (mask shared with unified_transformer.prefix_lm_mask).

The 2-D ids derive from ``prefix_len`` ([B]) and the absolute index, so the
decode cache stays (k, v) and token-by-token decode is exact.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ..configuration_utils import PretrainedConfig
from ..model_utils import PretrainedModel
from ..unified_transformer.modeling import prefix_lm_mask

__all__ = ["ChatGLMConfig", "ChatGLMModel", "ChatGLMForCausalLM"]


class ChatGLMConfig(PretrainedConfig):
    model_type = "chatglm"

    def __init__(self, vocab_size=130528, hidden_size=4096,
                 num_hidden_layers=28, num_attention_heads=32,
                 inner_hidden_size=16384, layernorm_epsilon=1e-5,
                 max_sequence_length=2048, initializer_range=0.02,
                 bos_token_id=130004, eos_token_id=130005,
                 pad_token_id=3, mask_token_id=130000, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.hidden_size = hidden_size
        self.num_hidden_layers = num_hidden_layers
        self.num_attention_heads = num_attention_heads
        self.inner_hidden_size = inner_hidden_size
        self.layernorm_epsilon = layernorm_epsilon
        self.max_sequence_length = max_sequence_length
        self.initializer_range = initializer_range
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.pad_token_id = pad_token_id
        self.mask_token_id = mask_token_id

    @property
    def head_dim(self):
        return self.hidden_size // self.num_attention_heads


def glm_2d_positions(prefix_len: torch.Tensor, past_len: int, S: int,
                     device) -> tuple:
    """(pos_ids, block_ids), each [B, S], for absolute indices
    past_len..past_len+S-1: context token i -> (i, 0); generated token
    i >= prefix -> (prefix - 1, i - prefix + 1)."""
    idx = torch.arange(past_len, past_len + S, device=device)[None, :]
    p = prefix_len.to(device)[:, None]
    pos = torch.minimum(idx, (p - 1).clamp(min=0))
    block = (idx - p + 1).clamp(min=0)
    return pos, block


def _rotate_half_pairs(x, cos, sin):
    """NeoX-style rotary over x[..., :dim] with per-token cos/sin [B,S,d/2]."""
    d = cos.shape[-1]
    x1, x2 = x[..., :d], x[..., d:2 * d]
    c = cos[:, :, None, :]
    s = sin[:, :, None, :]
    return torch.cat([x1 * c - x2 * s, x2 * c + x1 * s, x[..., 2 * d:]], -1)


class ChatGLMAttention(nn.Module):
    def __init__(self, config: ChatGLMConfig):
        super().__init__()
        h = config.hidden_size
        self.num_heads = config.num_attention_heads
        self.head_dim = config.head_dim
        # rotary over half the head dims, split again into two 2-D channels
        self.rot = self.head_dim // 2
        self.query_key_value = nn.Linear(h, 3 * h)
        self.dense = nn.Linear(h, h)

    def _cos_sin(self, ids):
        """ids: [B, S] -> cos/sin [B, S, rot/2]."""
        n = self.rot // 2
        inv = 1.0 / (10000.0 ** (torch.arange(n, device=ids.device).float() * 2
                                 / self.rot))
        freqs = ids.float()[..., None] * inv  # [B, S, n]
        return freqs.cos(), freqs.sin()

    def forward(self, x, mask, pos_ids, block_ids, past_key_value=None,
                use_cache=False):
        B, S, H = x.shape
        qkv = self.query_key_value(x)
        q, k, v = qkv.chunk(3, dim=-1)
        shape = (B, S, self.num_heads, self.head_dim)
        q, k, v = q.view(shape), k.view(shape), v.view(shape)
        # first rot dims rotate with pos ids, next rot dims with block ids
        pcos, psin = self._cos_sin(pos_ids)
        bcos, bsin = self._cos_sin(block_ids)
        q1 = _rotate_half_pairs(q[..., :self.rot], pcos.to(q.dtype), psin.to(q.dtype))
        k1 = _rotate_half_pairs(k[..., :self.rot], pcos.to(q.dtype), psin.to(q.dtype))
        q2 = _rotate_half_pairs(q[..., self.rot:], bcos.to(q.dtype), bsin.to(q.dtype))
        k2 = _rotate_half_pairs(k[..., self.rot:], bcos.to(q.dtype), bsin.to(q.dtype))
        q = torch.cat([q1, q2], dim=-1).transpose(1, 2)
        k = torch.cat([k1, k2], dim=-1).transpose(1, 2)
        v = v.transpose(1, 2)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=2)
            v = torch.cat([past_key_value[1], v], dim=2)
        present = (k, v) if use_cache else None
        out = F.scaled_dot_product_attention(q, k, v, attn_mask=mask)
        out = self.dense(out.transpose(1, 2).reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class ChatGLMBlock(nn.Module):
    def __init__(self, config: ChatGLMConfig):
        super().__init__()
        h = config.hidden_size
        self.input_layernorm = nn.LayerNorm(h, eps=config.layernorm_epsilon)
        self.attention = ChatGLMAttention(config)
        self.post_attention_layernorm = nn.LayerNorm(
            h, eps=config.layernorm_epsilon)
        self.dense_h_to_4h = nn.Linear(h, config.inner_hidden_size)
        self.dense_4h_to_h = nn.Linear(config.inner_hidden_size, h)
        self.alpha = (2 * config.num_hidden_layers) ** 0.5

    def forward(self, x, mask, pos_ids, block_ids, past_key_value=None,
                use_cache=False):
        ln1 = self.input_layernorm(x)
        attn = self.attention(ln1, mask, pos_ids, block_ids,
                              past_key_value, use_cache)
        if use_cache:
            attn, present = attn
        # GLM residual: alpha * LN(x) + sublayer(LN(x))
        x = self.alpha * ln1 + attn
        ln2 = self.post_attention_layernorm(x)
        mlp = self.dense_4h_to_h(F.gelu(self.dense_h_to_4h(ln2)))
        x = self.alpha * ln2 + mlp
        if use_cache:
            return x, present
        return x


class ChatGLMPretrainedModel(PretrainedModel):
    config_class = ChatGLMConfig
    base_model_prefix = "chatglm"

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


class ChatGLMModel(ChatGLMPretrainedModel):
    def __init__(self, config: ChatGLMConfig):
        super().__init__(config)
        self.word_embeddings = nn.Embedding(config.vocab_size,
                                            config.hidden_size)
        self.layers = nn.ModuleList(
            [ChatGLMBlock(config) for _ in range(config.num_hidden_layers)])
        self.final_layernorm = nn.LayerNorm(config.hidden_size,
                                            eps=config.layernorm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.word_embeddings

    def forward(self, input_ids=None, prefix_len=None, past_key_values=None,
                use_cache=False, inputs_embeds=None):
        x = (self.word_embeddings(input_ids) if inputs_embeds is None
             else inputs_embeds)
        B, S = x.shape[:2]
        device = x.device
        past_len = 0
        if past_key_values is not None and past_key_values[0] is not None:
            past_len = past_key_values[0][0].shape[2]
        if prefix_len is None:
            prefix_len = torch.full((B,), past_len + S if past_len == 0 else 0,
                                    dtype=torch.long, device=device)
        pos_ids, block_ids = glm_2d_positions(prefix_len, past_len, S,
                                              device)
        mask = prefix_lm_mask(S, prefix_len, past_len, device)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            past = past_key_values[i] if past_key_values is not None else None
            out = layer(x, mask, pos_ids, block_ids, past, use_cache)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.final_layernorm(x)
        if use_cache:
            return x, presents
        return x


class ChatGLMForCausalLM(ChatGLMPretrainedModel):
    def __init__(self, config: ChatGLMConfig):
        super().__init__(config)
        self.chatglm = ChatGLMModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size,
                                 bias=False)
        self.init_weights()

    def get_input_embeddings(self):
        return self.chatglm.word_embeddings

    def forward(self, input_ids=None, prefix_len=None, labels=None,
                past_key_values=None, use_cache=False, inputs_embeds=None,
                **kwargs):
        out = self.chatglm(input_ids, prefix_len, past_key_values, use_cache,
                           inputs_embeds=inputs_embeds)
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
    def generate(self, input_ids, max_new_tokens=32, do_sample=False,
                 temperature=1.0, eos_token_id=None, **kwargs):
        eos = eos_token_id if eos_token_id is not None \
            else self.config.eos_token_id
        B, S = input_ids.shape
        device = input_ids.device
        prefix_len = torch.full((B,), S, dtype=torch.long, device=device)
        logits, past = self.forward(input_ids, prefix_len, use_cache=True)
        cur = logits[:, -1].float()
        unfinished = torch.ones(B, dtype=torch.bool, device=device)
        tokens = []
        for _ in range(max_new_tokens):
            if do_sample:
                token = torch.multinomial(
                    (cur / max(temperature, 1e-6)).softmax(-1), 1).squeeze(-1)
            else:
                token = cur.argmax(-1)
            token = torch.where(unfinished, token,
                                torch.full_like(token, self.config.pad_token_id))
            tokens.append(token)
            unfinished = unfinished & (token != eos)
            if not unfinished.any():
                break
            logits, past = self.forward(token[:, None], prefix_len,
                                        past_key_values=past, use_cache=True)
            cur = logits[:, -1].float()
        return torch.stack(tokens, dim=1), None
