"""Checkpoint name/format conversion (HF-transformers -> paddlenlp_amd).

Reference behavior: paddlenlp/transformers/conversion_utils.py
(StateDictNameMapping :677, ConversionMixin :1134 — converts HF torch
checkpoints into the framework's naming/fusion layout; per-family mappings
from `_get_name_mappings`, llama/modeling.py:1243).

This framework already stores HF-compatible safetensors; conversion here is
name remapping (model. -> llama. etc.) plus qkv / gate_up fusion so stock
HuggingFace Llama/Qwen2/Mistral checkpoints load directly.
"""
from __future__ import annotations

import os
from typing import Dict

import torch

from ..utils.log import logger


def _fuse_llama_family(sd: Dict[str, torch.Tensor], prefix: str, config) -> Dict[str, torch.Tensor]:
    out = {}
    n_layers = config.num_hidden_layers
    fuse_qkv = getattr(config, "fuse_attention_qkv", True)
    fuse_ffn = getattr(config, "fuse_attention_ffn", True)
    consumed = set()
    for i in range(n_layers):
        a = f"{prefix}.layers.{i}.self_attn."
        m = f"{prefix}.layers.{i}.mlp."
        if fuse_qkv and a + "q_proj.weight" in sd:
            out[a + "qkv_proj.weight"] = torch.cat(
                [sd[a + "q_proj.weight"], sd[a + "k_proj.weight"], sd[a + "v_proj.weight"]], dim=0)
            consumed |= {a + "q_proj.weight", a + "k_proj.weight", a + "v_proj.weight"}
            if a + "q_proj.bias" in sd:
                out[a + "qkv_proj.bias"] = torch.cat(
                    [sd[a + "q_proj.bias"], sd[a + "k_proj.bias"], sd[a + "v_proj.bias"]], dim=0)
                consumed |= {a + "q_proj.bias", a + "k_proj.bias", a + "v_proj.bias"}
        if fuse_ffn and m + "gate_proj.weight" in sd:
            out[m + "gate_up_fused_proj.weight"] = torch.cat(
                [sd[m + "gate_proj.weight"], sd[m + "up_proj.weight"]], dim=0)
            consumed |= {m + "gate_proj.weight", m + "up_proj.weight"}
    for k, v in sd.items():
        if k not in consumed:
            out[k] = v
    return out


def convert_hf_state_dict(sd: Dict[str, torch.Tensor], config) -> Dict[str, torch.Tensor]:
    """HF names (model.layers.N...) -> framework names + fused projections."""
    model_type = config.model_type
    base_prefix = {"llama": "llama", "qwen2": "qwen2", "mistral": "mistral",
                   "mixtral": "mixtral"}.get(model_type, model_type)
    renamed = {}
    for k, v in sd.items():
        nk = k
        if nk.startswith("model."):
            nk = base_prefix + nk[len("model"):]
        renamed[nk] = v
    if model_type in ("llama", "qwen2", "mistral"):
        renamed = _fuse_llama_family(renamed, base_prefix, config)
    return renamed


def convert_hf_checkpoint(hf_dir: str, out_dir: str, config=None) -> None:
    """Convert an HF-format local checkpoint directory in one pass."""
    from safetensors import safe_open
    from safetensors.torch import save_file

    from .auto.configuration import AutoConfig

    config = config or AutoConfig.from_pretrained(hf_dir)
    sd = {}
    files = [f for f in sorted(os.listdir(hf_dir))
             if f.startswith("model") and f.endswith(".safetensors")]
    for fname in files:
        with safe_open(os.path.join(hf_dir, fname), framework="pt", device="cpu") as f:
            for k in f.keys():
                sd[k] = f.get_tensor(k)
    converted = convert_hf_state_dict(sd, config)
    os.makedirs(out_dir, exist_ok=True)
    config.save_pretrained(out_dir)
    save_file({k: v.contiguous() for k, v in converted.items()},
              os.path.join(out_dir, "model.safetensors"), metadata={"format": "pt"})
    logger.info(f"Converted {len(converted)} tensors from {hf_dir} -> {out_dir}")
