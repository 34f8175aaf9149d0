"""Mamba (selective state-space) model family
(reference: paddlenlp/transformers/mamba/modeling.py).

Block: in_proj -> [x | z]; causal depthwise conv1d + SiLU on x; x_proj ->
(dt, B, C); selective scan h_t = exp(dt*A) h_{t-1} + dt*B x_t, y = C h + D x;
gated by SiLU(z); out_proj.  The scan here is a plain sequential recurrence
in fp32 (correctness oracle; the CDNA4 chunked parallel-scan kernel is the
planned serving path).  Decode caches (conv window, ssm state) give O(1)
per-token stepping.
"""
from __future__ import annotations

import math

import torch
import torch.nn as nn
import torch.nn.functional as F

from ... import ops
from ...generation import GenerationConfig, GenerationMixin
from ..llama.modeling import LlamaRMSNorm
from ..model_utils import PretrainedModel
from .configuration import MambaConfig

__all__ = ["MambaModel", "MambaForCausalLM", "MambaMixer"]


class MambaMixer(nn.Module):
    def __init__(self, config: MambaConfig):
        super().__init__()
        c = config
        h, d, n = c.hidden_size, c.intermediate_size, c.state_size
        self.d_inner = d
        self.d_state = n
        self.dt_rank = c.time_step_rank
        self.conv_kernel = c.conv_kernel

        self.in_proj = nn.Linear(h, 2 * d, bias=c.use_bias)
        self.conv1d = nn.Conv1d(d, d, c.conv_kernel, groups=d,
                                padding=c.conv_kernel - 1, bias=c.use_conv_bias)
        self.x_proj = nn.Linear(d, self.dt_rank + 2 * n, bias=False)
        self.dt_proj = nn.Linear(self.dt_rank, d, bias=True)
        # S4D-real initialization: A = -[1..n] per channel
        A = torch.arange(1, n + 1, dtype=torch.float32).repeat(d, 1)
        self.A_log = nn.Parameter(torch.log(A))
        self.D = nn.Parameter(torch.ones(d))
        self.out_proj = nn.Linear(d, h, bias=c.use_bias)

        # dt bias init so softplus(dt) starts in [1e-3, 1e-1]
        dt = torch.exp(torch.rand(d) * (math.log(0.1) - math.log(1e-3))
                       + math.log(1e-3)).clamp(min=1e-4)
        with torch.no_grad():
            self.dt_proj.bias.copy_(dt + torch.log(-torch.expm1(-dt)))

    def forward(self, hidden, cache=None):
        """hidden [B, S, H]; cache = (conv_state [B, D, K-1],
        ssm_state [B, D, N]) for O(1) decode."""
        B, S, _ = hidden.shape
        d, n = self.d_inner, self.d_state
        xz = self.in_proj(hidden)                        # [B, S, 2D]
        x, z = xz.split([d, d], dim=-1)
        x = x.transpose(1, 2)                            # [B, D, S]

        if cache is not None:
            conv_state, ssm_state = cache
            x_all = torch.cat([conv_state, x], dim=2)    # [B, D, K-1+S]
            new_conv_state = x_all[:, :, -(self.conv_kernel - 1):]
            x = F.conv1d(x_all, self.conv1d.weight, self.conv1d.bias,
                         groups=d)[:, :, -S:]
        else:
            ssm_state = hidden.new_zeros(B, d, n)
            new_conv_state = F.pad(
                x, (self.conv_kernel - 1 - min(S, self.conv_kernel - 1), 0)
            )[:, :, -(self.conv_kernel - 1):]
            x = self.conv1d(x)[:, :, :S]
        x = F.silu(x).transpose(1, 2)                    # [B, S, D]

        dbc = self.x_proj(x)                             # [B, S, r+2N]
        dt, Bm, Cm = dbc.split([self.dt_rank, n, n], dim=-1)
        dt = F.softplus(self.dt_proj(dt)).float()        # [B, S, D]
        A = -torch.exp(self.A_log.float())               # [D, N]

        # sequential selective scan (fp32)
        hstate = ssm_state.float()                       # [B, D, N]
        xf = x.float()
        Bf = Bm.float()
        Cf = Cm.float()
        ys = []
        for t in range(S):
            dA = torch.exp(dt[:, t, :, None] * A[None])          # [B, D, N]
            dBx = dt[:, t, :, None] * Bf[:, t, None, :] * xf[:, t, :, None]
            hstate = hstate * dA + dBx
            ys.append((hstate * Cf[:, t, None, :]).sum(-1))      # [B, D]
        y = torch.stack(ys, dim=1)                               # [B, S, D]
        y = y + xf * self.D.float()[None, None]
        y = (y.to(hidden.dtype)) * F.silu(z)
        out = self.out_proj(y)
        return out, (new_conv_state, hstate.to(hidden.dtype))


class MambaBlock(nn.Module):
    def __init__(self, config: MambaConfig):
        super().__init__()
        self.norm = LlamaRMSNorm(config)
        self.mixer = MambaMixer(config)

    def forward(self, x, cache=None):
        out, new_cache = self.mixer(self.norm(x), cache)
        return x + out, new_cache


class MambaPretrainedModel(PretrainedModel):
    config_class = MambaConfig
    base_model_prefix = "mamba"

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)
        # linears keep torch default (mamba relies on its dt/A init)


class MambaModel(MambaPretrainedModel):
    def __init__(self, config: MambaConfig):
        super().__init__(config)
        self.embeddings = nn.Embedding(config.vocab_size, config.hidden_size)
        self.layers = nn.ModuleList(
            [MambaBlock(config) for _ in range(config.num_hidden_layers)])
        self.norm_f = LlamaRMSNorm(config)
        self.init_weights()

    def get_input_embeddings(self):
        return self.embeddings

    def forward(self, input_ids, past_key_values=None, use_cache=False):
        x = self.embeddings(input_ids)
        presents = [] if use_cache else None
        for i, layer in enumerate(self.layers):
            cache = past_key_values[i] if past_key_values is not None else None
            x, new_cache = layer(x, cache)
            if use_cache:
                presents.append(new_cache)
        x = self.norm_f(x)
        if use_cache:
            return x, presents
        return x


class MambaForCausalLM(MambaPretrainedModel, GenerationMixin):
    _tied_weights_keys = ["lm_head.weight"]

    def __init__(self, config: MambaConfig):
        super().__init__(config)
        self.mamba = MambaModel(config)
        self.lm_head = nn.Linear(config.hidden_size, config.vocab_size, bias=False)
        if config.tie_word_embeddings:
            self.lm_head.weight = self.mamba.embeddings.weight
        self.generation_config = GenerationConfig.from_model_config(config)

    def tie_weights(self):
        if self.config.tie_word_embeddings:
            self.lm_head.weight = self.mamba.embeddings.weight

    def get_input_embeddings(self):
        return self.mamba.embeddings

    def get_output_embeddings(self):
        return self.lm_head

    def forward(self, input_ids=None, labels=None, past_key_values=None,
                use_cache=False, attention_mask=None, **kwargs):
        out = self.mamba(input_ids, past_key_values, use_cache)
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
