"""Prefix tuning: learned past-key-values injected at every layer.

Reference behavior: paddlenlp/peft/prefix/prefix_model.py
(PrefixModelForCausalLM with past-key-value injection).
"""
from __future__ import annotations

import json
import os
from dataclasses import asdict, dataclass

import torch
import torch.nn as nn

PREFIX_CONFIG_NAME = "prefix_config.json"
PREFIX_WEIGHTS_NAME = "prefix_model_state.safetensors"


@dataclass
class PrefixConfig:
    num_prefix_tokens: int = 16
    prefix_projection: bool = True
    prefix_projection_hidden_size: int = 512

    def save_pretrained(self, path):
        os.makedirs(path, exist_ok=True)
        with open(os.path.join(path, PREFIX_CONFIG_NAME), "w") as f:
            json.dump(asdict(self), f, indent=2)

    @classmethod
    def from_pretrained(cls, path):
        with open(os.path.join(path, PREFIX_CONFIG_NAME)) as f:
            return cls(**json.load(f))


class PrefixEncoder(nn.Module):
    def __init__(self, cfg: PrefixConfig, num_layers: int, num_kv_heads: int, head_dim: int):
        super().__init__()
        self.cfg = cfg
        self.num_layers = num_layers
        self.num_kv_heads = num_kv_heads
        self.head_dim = head_dim
        out_dim = num_layers * 2 * num_kv_heads * head_dim
        if cfg.prefix_projection:
            self.embedding = nn.Embedding(cfg.num_prefix_tokens, cfg.prefix_projection_hidden_size)
            self.proj = nn.Sequential(
                nn.Linear(cfg.prefix_projection_hidden_size, cfg.prefix_projection_hidden_size),
                nn.Tanh(),
                nn.Linear(cfg.prefix_projection_hidden_size, out_dim),
            )
        else:
            self.embedding = nn.Embedding(cfg.num_prefix_tokens, out_dim)
            self.proj = nn.Identity()

    def forward(self, batch_size: int, device, dtype):
        idx = torch.arange(self.cfg.num_prefix_tokens, device=device)
        pkv = self.proj(self.embedding(idx)).to(dtype)  # [P, L*2*Hk*D]
        P = self.cfg.num_prefix_tokens
        pkv = pkv.view(P, self.num_layers, 2, self.num_kv_heads, self.head_dim)
        pkv = pkv.permute(1, 2, 0, 3, 4)  # [L, 2, P, Hk, D]
        out = []
        for l in range(self.num_layers):
            k = pkv[l, 0].unsqueeze(0).expand(batch_size, -1, -1, -1)  # [B, P, Hk, D]
            v = pkv[l, 1].unsqueeze(0).expand(batch_size, -1, -1, -1)
            out.append((k.contiguous(), v.contiguous()))
        return out


class PrefixModelForCausalLM(nn.Module):
    def __init__(self, model, prefix_config: PrefixConfig):
        super().__init__()
        self.model = model
        self.prefix_config = prefix_config
        cfg = model.config
        self.prefix_encoder = PrefixEncoder(
            prefix_config, cfg.num_hidden_layers,
            getattr(cfg, "num_key_value_heads", cfg.num_attention_heads), cfg.head_dim,
        )
        for p in self.model.parameters():
            p.requires_grad_(False)

    @property
    def config(self):
        return self.model.config

    def forward(self, input_ids=None, labels=None, **kwargs):
        B = input_ids.shape[0]
        dtype = next(self.model.parameters()).dtype
        pkv = self.prefix_encoder(B, input_ids.device, dtype)
        out = self.model(input_ids=input_ids, labels=labels,
                         past_key_values=pkv, use_cache=True, **kwargs)
        # strip the cache from the returned tuple for training parity
        if labels is not None:
            return out[0], out[1]
        return out[0]

    def save_pretrained(self, path: str):
        from safetensors.torch import save_file

        os.makedirs(path, exist_ok=True)
        self.prefix_config.save_pretrained(path)
        sd = {f"prefix_encoder.{k}": v.contiguous().cpu()
              for k, v in self.prefix_encoder.state_dict().items()}
        save_file(sd, os.path.join(path, PREFIX_WEIGHTS_NAME), metadata={"format": "pt"})

    @classmethod
    def from_pretrained(cls, model, path: str):
        from safetensors.torch import load_file

        cfg = PrefixConfig.from_pretrained(path)
        m = cls(model, cfg)
        sd = load_file(os.path.join(path, PREFIX_WEIGHTS_NAME))
        sd = {k.replace("prefix_encoder.", ""): v for k, v in sd.items()}
        m.prefix_encoder.load_state_dict(sd)
        return m
