"""model_type -> (module, class names) registry shared by the Auto* classes.

Reference: paddlenlp/transformers/auto/{configuration,modeling,tokenizer}.py
name->class registries resolved via importlib.
"""
import importlib
import json
import os

# model_type: (module path under paddlenlp_amd.transformers, config class,
#              causal-lm class, base-model class)
MODEL_REGISTRY = {
    "llama": ("llama", "LlamaConfig", "LlamaForCausalLM", "LlamaModel"),
    "gpt2": ("gpt", "GPTConfig", "GPTForCausalLM", "GPTModel"),
    "gpt": ("gpt", "GPTConfig", "GPTForCausalLM", "GPTModel"),
    "qwen2": ("qwen2", "Qwen2Config", "Qwen2ForCausalLM", "Qwen2Model"),
    "mixtral": ("mixtral", "MixtralConfig", "MixtralForCausalLM", "MixtralModel"),
    "qwen2_moe": ("qwen2_moe", "Qwen2MoeConfig", "Qwen2MoeForCausalLM", "Qwen2MoeModel"),
    "mistral": ("mistral", "MistralConfig", "MistralForCausalLM", "MistralModel"),
}


def resolve_model_type(path: str) -> str:
    config_file = os.path.join(path, "config.json") if os.path.isdir(path) else path
    with open(config_file) as f:
        cfg = json.load(f)
    model_type = cfg.get("model_type")
    if model_type is None:
        archs = cfg.get("architectures") or []
        for arch in archs:
            for mt, (_, _, lm_cls, base_cls) in MODEL_REGISTRY.items():
                if arch in (lm_cls, base_cls):
                    return mt
        raise ValueError(f"Cannot infer model_type from {config_file}")
    return model_type


def get_class(model_type: str, kind: str):
    """kind in {config, causal_lm, base}."""
    if model_type not in MODEL_REGISTRY:
        raise ValueError(
            f"Unknown model_type '{model_type}'. Registered: {sorted(MODEL_REGISTRY)}"
        )
    module_name, cfg_cls, lm_cls, base_cls = MODEL_REGISTRY[model_type]
    module = importlib.import_module(f"paddlenlp_amd.transformers.{module_name}")
    name = {"config": cfg_cls, "causal_lm": lm_cls, "base": base_cls}[kind]
    return getattr(module, name)
