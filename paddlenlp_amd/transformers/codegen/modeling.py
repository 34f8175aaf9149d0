"""CodeGen family (reference: paddlenlp/transformers/codegen/modeling.py).

Salesforce CodeGen: GPT-J-shaped decoder (parallel residual, one pre-LN per
block, interleaved-pair rotary on the first ``rotary_dim`` head dims) with a
FUSED no-bias qkv projection and a biased LM head.  The reference stores the
fused qkv in an mp_num=4-blocked layout for TP checkpoint sharding; we use
the plain [q|k|v] layout (conversion handles re-blocking when importing
checkpoints) — behavior is identical.
"""
from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..configuration_utils import PretrainedConfig
from ..gptj.modeling import _gptj_rope
from ..model_utils import PretrainedModel

__all__ = ["CodeGenConfig", "CodeGenModel", "CodeGenForCausalLM"]


class CodeGenConfig(PretrainedConfig):
    model_type = "codegen"

    def __init__(self, vocab_size=50400, n_embd=1024, n_layer=20, n_head=16,
                 n_inner=None, rotary_dim=32, layer_norm_epsilon=1e-5,
                 max_position_embeddings=2048, initializer_range=0.02,
                 rope_theta=10000.0, bos_token_id=1, eos_token_id=50256,
                 pad_token_id=None, **kwargs):
        super().__init__(**kwargs)
        self.vocab_size = vocab_size
        self.n_embd = n_embd
        self.n_layer = n_layer
        self.n_head = n_head
        self.n_inner = n_inner if n_inner is not None else 4 * n_embd
        self.rotary_dim = rotary_dim
        self.layer_norm_epsilon = layer_norm_epsilon
        self.max_position_embeddings = max_position_embeddings
        self.initializer_range = initializer_range
        self.rope_theta = rope_theta
        self.bos_token_id = bos_token_id
        self.eos_token_id = eos_token_id
        self.pad_token_id = pad_token_id

    # canonical-name aliases used by shared infra
    attribute_map = {
        "hidden_size": "n_embd",
        "num_hidden_layers": "n_layer",
        "num_attention_heads": "n_head",
        "intermediate_size": "n_inner",
    }

    @property
    def head_dim(self):
        return self.n_embd // self.n_head


class CodeGenAttention(nn.Module):
    def __init__(self, config: CodeGenConfig):
        super().__init__()
        h = config.n_embd
        self.num_heads = config.n_head
        self.head_dim = config.head_dim
        self.rotary_dim = config.rotary_dim
        self.rope_theta = config.rope_theta
        self.qkv_proj = nn.Linear(h, 3 * h, bias=False)
        self.out_proj = nn.Linear(h, h, bias=False)

    def _cos_sin(self, S, device, dtype, offset):
        n = self.rotary_dim // 2
        inv = 1.0 / (self.rope_theta
                     ** (torch.arange(n, device=device).float() * 2 / self.rotary_dim))
        t = torch.arange(offset, offset + S, device=device).float()
        freqs = torch.outer(t, inv)
        return freqs.cos().to(dtype), freqs.sin().to(dtype)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        B, S, H = x.shape
        shape = (B, S, self.num_heads, self.head_dim)
        q, k, v = self.qkv_proj(x).chunk(3, dim=-1)
        q, k, v = q.view(shape), k.view(shape), v.view(shape)
        cos, sin = self._cos_sin(S, x.device, x.dtype, position_offset)
        q = _gptj_rope(q, cos, sin, self.rotary_dim)
        k = _gptj_rope(k, cos, sin, self.rotary_dim)
        if past_key_value is not None:
            k = torch.cat([past_key_value[0], k], dim=1)
            v = torch.cat([past_key_value[1], v], dim=1)
        present = (k, v) if use_cache else None
        out = ops.flash_attention(q, k, v, causal=True)
        out = self.out_proj(out.reshape(B, S, H))
        if use_cache:
            return out, present
        return out


class CodeGenBlock(nn.Module):
    def __init__(self, config: CodeGenConfig):
        super().__init__()
        h = config.n_embd
        self.ln_1 = nn.LayerNorm(h, eps=config.layer_norm_epsilon)
        self.attn = CodeGenAttention(config)
        self.fc_in = nn.Linear(h, config.n_inner)
        self.fc_out = nn.Linear(config.n_inner, h)

    def forward(self, x, past_key_value=None, use_cache=False, position_offset=0):
        ln = self.ln_1(x)
        attn = self.attn(ln, past_key_value, use_cache, position_offset)
        if use_cache:
            attn, present = attn
        mlp = self.fc_out(F.gelu(self.fc_in(ln), approximate="tanh"))
        x = x + attn + mlp
        if use_cache:
            return x, present
        return x


class CodeGenPretrainedModel(PretrainedModel):
    config_class = CodeGenConfig
    base_model_prefix = "transformer"

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


class CodeGenModel(CodeGenPretrainedModel):
    def __init__(self, config: CodeGenConfig):
        super().__init__(config)
        self.wte = nn.Embedding(config.vocab_size, config.n_embd)
        self.h = nn.ModuleList(
            [CodeGenBlock(config) for _ in range(config.n_layer)])
        self.ln_f = nn.LayerNorm(config.n_embd, eps=config.layer_norm_epsilon)
        self.init_weights()

    def get_input_embeddings(self):
        return self.wte

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.wte(input_ids)
        offset = 0
        if past_key_values is not None and past_key_values[0] is not None:
            offset = past_key_values[0][0].shape[1]
        presents = [] if use_cache else None
        for i, block in enumerate(self.h):
            past = past_key_values[i] if past_key_values is not None else None
            out = block(x, past, use_cache, offset)
            if use_cache:
                x, present = out
                presents.append(present)
            else:
                x = out
        x = self.ln_f(x)
        if use_cache:
            return x, presents
        return x


class CodeGenForCausalLM(CodeGenPretrainedModel, GenerationMixin):
    def __init__(self, config: CodeGenConfig):
        super().__init__(config)
        self.transformer = CodeGenModel(config)
        self.lm_head = nn.Linear(config.n_embd, config.vocab_size, bias=True)
        self.init_weights()
        self.generation_config = GenerationConfig.from_model_config(config)

    def get_input_embeddings(self):
        return self.transformer.wte

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.transformer(input_ids, past_key_values, use_cache)
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
