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


def _convert_bert(sd: Dict[str, torch.Tensor], config) -> Dict[str, torch.Tensor]:
    """HF BERT naming -> shared-encoder-core naming + fused qkv."""
    import re

    out = {}
    qkv: Dict[str, dict] = {}
    rules = [
        (r"\.embeddings\.LayerNorm\.", ".embeddings.layer_norm."),
        (r"\.encoder\.layer\.(\d+)\.attention\.output\.dense\.",
         r".encoder.layers.\1.self_attn.out_proj."),
        (r"\.encoder\.layer\.(\d+)\.attention\.output\.LayerNorm\.",
         r".encoder.layers.\1.attn_norm."),
        (r"\.encoder\.layer\.(\d+)\.intermediate\.dense\.",
         r".encoder.layers.\1.fc_in."),
        (r"\.encoder\.layer\.(\d+)\.output\.dense\.",
         r".encoder.layers.\1.fc_out."),
        (r"\.encoder\.layer\.(\d+)\.output\.LayerNorm\.",
         r".encoder.layers.\1.mlp_norm."),
        (r"cls\.predictions\.transform\.dense\.", "cls.dense."),
        (r"cls\.predictions\.transform\.LayerNorm\.", "cls.layer_norm."),
        (r"cls\.predictions\.decoder\.", "cls.decoder."),
        (r"cls\.predictions\.bias$", "cls.decoder.bias"),
    ]
    for k, v in sd.items():
        m = re.match(r"(.*\.encoder\.layer\.(\d+))\.attention\.self\.(query|key|value)\.(weight|bias)$", k)
        if m:
            layer_key = m.group(1).replace(".layer.", ".layers.") \
                + ".self_attn.qkv_proj." + m.group(4)
            qkv.setdefault(layer_key, {})[m.group(3)] = v
            continue
        nk = k
        for pat, rep in rules:
            nk = re.sub(pat, rep, nk)
        out[nk] = v
    for key, parts in qkv.items():
        out[key] = torch.cat([parts["query"], parts["key"], parts["value"]], dim=0)
    return out


def _convert_t5(sd: Dict[str, torch.Tensor], config) -> Dict[str, torch.Tensor]:
    """HF T5 naming -> this framework's stack naming."""
    import re

    rules = [
        (r"\.block\.(\d+)\.layer\.0\.SelfAttention\.", r".blocks.\1.self_attn."),
        (r"\.block\.(\d+)\.layer\.0\.layer_norm\.", r".blocks.\1.self_norm."),
        (r"\.block\.(\d+)\.layer\.1\.EncDecAttention\.", r".blocks.\1.cross_attn."),
        (r"decoder\.block\.(\d+)\.layer\.1\.layer_norm\.", r"decoder.blocks.\1.cross_norm."),
        (r"decoder\.block\.(\d+)\.layer\.2\.DenseReluDense\.", r"decoder.blocks.\1.ff."),
        (r"decoder\.block\.(\d+)\.layer\.2\.layer_norm\.", r"decoder.blocks.\1.ff_norm."),
        (r"encoder\.block\.(\d+)\.layer\.1\.DenseReluDense\.", r"encoder.blocks.\1.ff."),
        (r"encoder\.block\.(\d+)\.layer\.1\.layer_norm\.", r"encoder.blocks.\1.ff_norm."),
        (r"\.final_layer_norm\.", ".final_norm."),
    ]
    out = {}
    for k, v in sd.items():
        nk = k
        for pat, rep in rules:
            nk = re.sub(pat, rep, nk)
        out[nk] = v
    return out


def _unblock_codegen_qkv(w: torch.Tensor, mp_num: int = 4) -> torch.Tensor:
    """Re-block a CodeGen fused qkv weight into plain [q|k|v] rows.

    HF/reference CodeGen stores qkv_proj as mp_num blocks, each laid out
    [q_i; v_i; k_i] (codegen/modeling.py:160-164 splits in q, v, k order).
    Returns the [3h, h] weight ordered [q; k; v] for this framework's
    straightforward chunk(3).
    """
    three_h, h = w.shape
    part = three_h // mp_num          # rows per block
    piece = part // 3                 # rows per tensor within a block
    blocks = w.view(mp_num, 3, piece, h)
    q = blocks[:, 0].reshape(-1, h)
    v = blocks[:, 1].reshape(-1, h)
    k = blocks[:, 2].reshape(-1, h)
    return torch.cat([q, k, v], dim=0)


def _convert_codegen(sd: Dict[str, torch.Tensor], config) -> Dict[str, torch.Tensor]:
    out = {}
    for k, v in sd.items():
        nk = k
        if nk.endswith("attn.qkv_proj.weight"):
            v = _unblock_codegen_qkv(v)
        out[nk] = v
    return out


def convert_hf_state_dict(sd: Dict[str, torch.Tensor], config) -> Dict[str, torch.Tensor]:
    """HF names (model.layers.N...) -> framework names + fused projections."""
    model_type = config.model_type
    if model_type == "bert":
        return _convert_bert(sd, config)
    if model_type == "codegen":
        return _convert_codegen(sd, config)
    if model_type == "t5":
        converted = _convert_t5(sd, config)
        # T5ForConditionalGeneration nests the stacks under "t5."
        return {("t5." + k if k.startswith(("encoder.", "decoder.", "shared."))
                 else k): v for k, v in converted.items()}
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
