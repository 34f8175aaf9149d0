"""PretrainedModel base class: from_pretrained / save_pretrained on safetensors.

Reference behavior: paddlenlp/transformers/model_utils.py:2161 (from_pretrained:
resolve config -> dtype -> TP-shard -> shard-by-shard load) and the sharded
safetensors save with `model.safetensors.index.json`.  This implementation is
torch-native: state dicts are plain torch tensors, files are safetensors, and
tensor-parallel split/merge happens through the `_get_tensor_parallel_mappings`
hook each model family implements (reference:
paddlenlp/transformers/conversion_utils.py:1134 ConversionMixin).
"""
from __future__ import annotations

import gc
import json
import os
import re
from typing import Dict, Optional

import torch
import torch.nn as nn

from ..utils.env import SAFE_WEIGHTS_INDEX_NAME, SAFE_WEIGHTS_NAME
from ..utils.log import logger
from .configuration_utils import PretrainedConfig

__all__ = ["PretrainedModel", "unwrap_model"]

_MAX_SHARD_SIZE = 5 * 1024**3  # 5 GB per safetensors shard


def unwrap_model(model: nn.Module) -> nn.Module:
    """Unwrap DDP/compiled wrappers down to the bare model."""
    while hasattr(model, "module") and isinstance(model.module, nn.Module):
        model = model.module
    if hasattr(model, "_orig_mod"):
        model = model._orig_mod
    return model


def dtype_byte_size(dtype: torch.dtype) -> float:
    if dtype == torch.bool:
        return 1 / 8
    bit_search = re.search(r"[^\d](\d+)(_.*)?$", str(dtype))
    if bit_search is None:
        raise ValueError(f"`dtype` is not a valid dtype: {dtype}.")
    return int(bit_search.groups()[0]) / 8


def shard_state_dict(state_dict: Dict[str, torch.Tensor], max_shard_size=_MAX_SHARD_SIZE):
    """Split a state dict into shards below max_shard_size.

    Returns ({filename: sub_state_dict}, index_dict_or_None).
    """
    sharded, current, current_size = [], {}, 0
    for key, weight in state_dict.items():
        weight_size = weight.numel() * dtype_byte_size(weight.dtype)
        if current and current_size + weight_size > max_shard_size:
            sharded.append(current)
            current, current_size = {}, 0
        current[key] = weight
        current_size += weight_size
    if current:
        sharded.append(current)

    if len(sharded) == 1:
        return {SAFE_WEIGHTS_NAME: sharded[0]}, None

    weight_map, shards = {}, {}
    for idx, shard in enumerate(sharded):
        name = SAFE_WEIGHTS_NAME.replace(
            ".safetensors", f"-{idx + 1:05d}-of-{len(sharded):05d}.safetensors"
        )
        shards[name] = shard
        for key in shard:
            weight_map[key] = name
    total = sum(w.numel() * dtype_byte_size(w.dtype) for w in state_dict.values())
    index = {"metadata": {"total_size": int(total)}, "weight_map": weight_map}
    return shards, index


class PretrainedModel(nn.Module):
    """Base class for every model family.

    Subclasses set `config_class`, `base_model_prefix`, and (for TP)
    implement `_get_tensor_parallel_mappings(config, is_split)` returning
    {param_name: split_fn_or_merge_fn}.
    """

    config_class = PretrainedConfig
    base_model_prefix = ""
    _keys_to_ignore_on_save = []
    # names tied together (e.g. lm_head.weight -> embed_tokens.weight)
    _tied_weights_keys: list = []

    def __init__(self, config: PretrainedConfig, *args, **kwargs):
        super().__init__()
        self.config = config

    # ------------------------------------------------------------------
    # init
    # ------------------------------------------------------------------
    def init_weights(self):
        """Apply `_init_weights` to every submodule (post-construction)."""
        if getattr(self.config, "_fast_init", True):
            self.apply(self._init_weights_wrapper)

    def _init_weights_wrapper(self, module):
        self._init_weights(module)

    def _init_weights(self, module):
        """Default truncated-normal init; per-family override."""
        std = getattr(self.config, "initializer_range", 0.02)
        if isinstance(module, nn.Linear):
            module.weight.data.normal_(mean=0.0, std=std)
            if module.bias is not None:
                module.bias.data.zero_()
        elif isinstance(module, nn.Embedding):
            module.weight.data.normal_(mean=0.0, std=std)

    def tie_weights(self):
        if getattr(self.config, "tie_word_embeddings", False):
            output_embeddings = self.get_output_embeddings()
            input_embeddings = self.get_input_embeddings()
            if output_embeddings is not None and input_embeddings is not None:
                output_embeddings.weight = input_embeddings.weight

    def get_input_embeddings(self) -> Optional[nn.Module]:
        base_model = getattr(self, self.base_model_prefix, None)
        if base_model is not None and base_model is not self:
            return base_model.get_input_embeddings()
        return None

    def get_output_embeddings(self) -> Optional[nn.Module]:
        return None

    # ------------------------------------------------------------------
    # hooks each family overrides for parallelism / HF-name conversion
    # (reference: llama/modeling.py:1243 _get_name_mappings,
    #  :1277 _get_tensor_parallel_mappings)
    # ------------------------------------------------------------------
    @classmethod
    def _get_tensor_parallel_mappings(cls, config, is_split=True):
        return {}

    # ------------------------------------------------------------------
    # save / load
    # ------------------------------------------------------------------
    def save_pretrained(self, save_directory: str, state_dict=None, max_shard_size=_MAX_SHARD_SIZE):
        from safetensors.torch import save_file

        os.makedirs(save_directory, exist_ok=True)
        model = unwrap_model(self)
        model.config.save_pretrained(save_directory)

        if state_dict is None:
            state_dict = model.state_dict()
        state_dict = {
            k: v for k, v in state_dict.items() if k not in self._keys_to_ignore_on_save
        }
        # drop tied duplicates: safetensors rejects shared storage
        for tied_key in self._tied_weights_keys:
            if tied_key in state_dict:
                del state_dict[tied_key]
        state_dict = {k: v.contiguous().cpu() for k, v in state_dict.items()}

        shards, index = shard_state_dict(state_dict, max_shard_size)
        # clear stale shards from a previous save
        for fname in os.listdir(save_directory):
            if fname.startswith("model") and fname.endswith(".safetensors"):
                os.remove(os.path.join(save_directory, fname))
        for name, shard in shards.items():
            save_file(shard, os.path.join(save_directory, name), metadata={"format": "pt"})
        if index is not None:
            with open(os.path.join(save_directory, SAFE_WEIGHTS_INDEX_NAME), "w") as f:
                json.dump(index, f, indent=2)
        logger.info(f"Model weights saved in {save_directory}")

    @classmethod
    def from_config(cls, config: PretrainedConfig, dtype=None, device=None, **kwargs):
        """Build a randomly initialized model from a config."""
        if dtype is None:
            dtype = getattr(config, "dtype", None) or "float32"
        if isinstance(dtype, str):
            dtype = getattr(torch, dtype)
        old_dtype = torch.get_default_dtype()
        try:
            torch.set_default_dtype(dtype)
            if device is not None:
                with torch.device(device):
                    model = cls(config, **kwargs)
            else:
                model = cls(config, **kwargs)
        finally:
            torch.set_default_dtype(old_dtype)
        model.init_weights()
        model.tie_weights()
        return model

    @classmethod
    def from_pretrained(cls, pretrained_model_name_or_path, config=None, dtype=None,
                        low_cpu_mem_usage=True, **kwargs):
        """Load from a local directory of config.json + safetensors shards.

        Resolution order (reference model_utils.py:2161): config -> dtype ->
        TP split -> shard-by-shard load with mismatch reporting.  Only local
        paths are supported (no network in this environment).
        """
        from safetensors import safe_open

        model_path = pretrained_model_name_or_path
        if config is None:
            config = cls.config_class.from_pretrained(model_path, **kwargs)
        if dtype is None:
            dtype = getattr(config, "dtype", None) or "float32"
        if isinstance(dtype, str):
            dtype = getattr(torch, dtype)

        # build the skeleton on meta device, then materialize as we load
        old_dtype = torch.get_default_dtype()
        try:
            torch.set_default_dtype(dtype)
            if low_cpu_mem_usage:
                with torch.device("meta"):
                    model = cls(config)
            else:
                model = cls(config)
        finally:
            torch.set_default_dtype(old_dtype)

        # resolve weight files
        index_file = os.path.join(model_path, SAFE_WEIGHTS_INDEX_NAME)
        single_file = os.path.join(model_path, SAFE_WEIGHTS_NAME)
        if os.path.isfile(index_file):
            with open(index_file) as f:
                index = json.load(f)
            shard_files = sorted(set(index["weight_map"].values()))
        elif os.path.isfile(single_file):
            shard_files = [SAFE_WEIGHTS_NAME]
        else:
            # fall back to any model*.safetensors shards present
            shard_files = sorted(
                f for f in os.listdir(model_path)
                if f.startswith("model") and f.endswith(".safetensors")
            )
            if not shard_files:
                raise FileNotFoundError(f"No safetensors weights found in {model_path}")

        tp_degree = getattr(config, "tensor_parallel_degree", 1)
        tp_actions = cls._get_tensor_parallel_mappings(config, is_split=True) if tp_degree > 1 else {}

        expected_keys = set(model.state_dict().keys())
        tied_keys = set(cls._tied_weights_keys)
        loaded_keys = set()
        for shard_file in shard_files:
            with safe_open(os.path.join(model_path, shard_file), framework="pt", device="cpu") as f:
                for key in f.keys():
                    if key not in expected_keys:
                        continue
                    tensor = f.get_tensor(key)
                    if key in tp_actions:
                        tensor = tp_actions[key](tensor)
                    tensor = tensor.to(dtype)
                    _assign_param(model, key, tensor)
                    loaded_keys.add(key)
            gc.collect()

        missing = expected_keys - loaded_keys - tied_keys
        # materialize any still-meta params (missing keys) with init
        if low_cpu_mem_usage:
            for name, param in list(model.named_parameters()) + list(model.named_buffers()):
                if param.is_meta and name not in tied_keys:
                    _assign_param(model, name, torch.empty(param.shape, dtype=param.dtype))
        if missing:
            logger.warning(f"Missing keys (left at init): {sorted(missing)}")
        model.tie_weights()
        model.eval()
        return model


# parallelism metadata attached to parameters that must survive reload
_PARAM_FLAGS = ("sequence_parallel", "is_column_parallel", "is_row_parallel", "no_sync")


def _assign_param(model: nn.Module, name: str, tensor: torch.Tensor):
    module = model
    parts = name.split(".")
    for p in parts[:-1]:
        module = getattr(module, p)
    leaf = parts[-1]
    old = getattr(module, leaf)
    if isinstance(old, nn.Parameter):
        new = nn.Parameter(tensor, requires_grad=old.requires_grad)
        for flag in _PARAM_FLAGS:
            if hasattr(old, flag):
                setattr(new, flag, getattr(old, flag))
        setattr(module, leaf, new)
    else:
        # buffer
        module.register_buffer(leaf, tensor, persistent=leaf in dict(module.named_buffers()))
