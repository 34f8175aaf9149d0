"""PretrainedConfig + LlmMetaConfig bridge.

Reference behavior: paddlenlp/transformers/configuration_utils.py:330
(PretrainedConfig with attribute_map) and :214-325 (LlmMetaConfig /
@llmmetaclass copying trainer args into the model config).  The MI355X
framework keeps the same single-source-of-truth pattern: the trainer's
parallel degrees and fusion flags are copied onto the config object so
modeling code reads one object.
"""
from __future__ import annotations

import copy
import json
import os
from dataclasses import dataclass
from typing import Any, Dict

from ..utils.env import CONFIG_NAME
from ..utils.log import logger


class PretrainedConfig:
    model_type: str = ""
    # maps legacy attribute names -> canonical names
    attribute_map: Dict[str, str] = {}

    def __init__(self, **kwargs):
        # generic defaults shared by every model config
        self.return_dict = kwargs.pop("return_dict", False)
        self.output_hidden_states = kwargs.pop("output_hidden_states", False)
        self.output_attentions = kwargs.pop("output_attentions", False)
        self.use_cache = kwargs.pop("use_cache", False)
        self.dtype = kwargs.pop("dtype", kwargs.pop("torch_dtype", None))
        self.tie_word_embeddings = kwargs.pop("tie_word_embeddings", False)

        # parallel-context fields (filled by LlmMetaConfig.set_llm_config)
        self.tensor_parallel_degree = kwargs.pop("tensor_parallel_degree", 1)
        self.tensor_parallel_rank = kwargs.pop("tensor_parallel_rank", 0)
        self.tensor_parallel_output = kwargs.pop("tensor_parallel_output", False)
        self.sequence_parallel = kwargs.pop("sequence_parallel", False)
        self.sep_parallel_degree = kwargs.pop("sep_parallel_degree", 1)
        self.context_parallel_degree = kwargs.pop("context_parallel_degree", 1)
        self.context_parallel_balanced = kwargs.pop("context_parallel_balanced", False)
        self.pipeline_parallel_degree = kwargs.pop("pipeline_parallel_degree", 1)
        self.recompute = kwargs.pop("recompute", False)
        self.recompute_granularity = kwargs.pop("recompute_granularity", "full")

        # op-fusion flags (reference: LlmMetaConfig fuse flags)
        self.use_flash_attention = kwargs.pop("use_flash_attention", True)
        self.use_fused_rms_norm = kwargs.pop("use_fused_rms_norm", True)
        self.use_fused_rope = kwargs.pop("use_fused_rope", True)
        self.use_fused_swiglu = kwargs.pop("use_fused_swiglu", True)
        self.use_fused_linear_cross_entropy = kwargs.pop("use_fused_linear_cross_entropy", False)

        for key, value in kwargs.items():
            try:
                setattr(self, key, value)
            except AttributeError:
                # read-only property (e.g. a derived `head_dim`) colliding
                # with a serialized field from a foreign (HF) config.json:
                # keep the derived value
                logger.warning(
                    f"Ignoring config field '{key}'={value}: read-only on "
                    f"{type(self).__name__}")

    # ---- attribute_map support (legacy-name aliasing) ----
    def __setattr__(self, key, value):
        if key != "attribute_map" and key in self.attribute_map:
            key = self.attribute_map[key]
        super().__setattr__(key, value)

    def __getattr__(self, key):
        if key != "attribute_map" and key in type(self).attribute_map:
            return getattr(self, type(self).attribute_map[key])
        raise AttributeError(
            f"'{type(self).__name__}' object has no attribute '{key}'"
        )

    # ---- serialization ----
    def to_dict(self) -> Dict[str, Any]:
        output = copy.deepcopy(self.__dict__)
        output["model_type"] = self.model_type
        if "_name_or_path" in output:
            del output["_name_or_path"]
        for k, v in list(output.items()):
            if hasattr(v, "to_dict"):
                output[k] = v.to_dict()
        return output

    def to_json_string(self) -> str:
        def _default(o):
            try:
                return str(o)
            except Exception:
                return None

        return json.dumps(self.to_dict(), indent=2, sort_keys=True, default=_default) + "\n"

    def save_pretrained(self, save_directory: str):
        os.makedirs(save_directory, exist_ok=True)
        path = os.path.join(save_directory, CONFIG_NAME)
        with open(path, "w", encoding="utf-8") as f:
            f.write(self.to_json_string())
        logger.info(f"Configuration saved in {path}")

    @classmethod
    def from_dict(cls, config_dict: Dict[str, Any], **kwargs) -> "PretrainedConfig":
        config_dict = dict(config_dict)
        config_dict.pop("model_type", None)
        # newer HF config.json nests rope under "rope_parameters"; promote
        # the fields so they beat subclass defaults
        rp = config_dict.get("rope_parameters")
        if isinstance(rp, dict):
            if "rope_theta" in rp and "rope_theta" not in config_dict:
                config_dict["rope_theta"] = rp["rope_theta"]
            if rp.get("rope_type", "default") != "default" and \
                    "rope_scaling" not in config_dict:
                config_dict["rope_scaling"] = {
                    k: v for k, v in rp.items() if k != "rope_theta"}
        config_dict.update(kwargs)
        return cls(**config_dict)

    @classmethod
    def from_pretrained(cls, pretrained_model_name_or_path: str, **kwargs) -> "PretrainedConfig":
        config_file = pretrained_model_name_or_path
        if os.path.isdir(config_file):
            config_file = os.path.join(config_file, CONFIG_NAME)
        if not os.path.isfile(config_file):
            raise FileNotFoundError(
                f"Config file not found at {pretrained_model_name_or_path} "
                f"(no network access: only local paths are supported)"
            )
        with open(config_file, "r", encoding="utf-8") as f:
            config_dict = json.load(f)
        return cls.from_dict(config_dict, **kwargs)

    def get(self, key, default=None):
        return getattr(self, key, default)

    def __repr__(self):
        return f"{type(self).__name__} {self.to_json_string()}"


@dataclass
class LlmMetaConfig:
    """Bridge copying trainer args onto the model config.

    Reference: paddlenlp/transformers/configuration_utils.py:230-325
    (`LlmMetaConfig.set_llm_config` called from llm/run_pretrain.py:415).
    """

    ATTRIBUTES = [
        # parallelism
        "tensor_parallel_degree",
        "tensor_parallel_rank",
        "tensor_parallel_output",
        "sequence_parallel",
        "sep_parallel_degree",
        "context_parallel_degree",
        "context_parallel_balanced",
        "pipeline_parallel_degree",
        # recompute
        "recompute",
        "recompute_granularity",
        # fusion flags
        "use_flash_attention",
        "use_fused_rms_norm",
        "use_fused_rope",
        "use_fused_swiglu",
        "use_fused_linear_cross_entropy",
    ]

    @classmethod
    def set_llm_config(cls, config: PretrainedConfig, args) -> None:
        for name in cls.ATTRIBUTES:
            if hasattr(args, name):
                setattr(config, name, getattr(args, name))
