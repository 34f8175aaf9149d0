"""AutoConfig (reference: paddlenlp/transformers/auto/configuration.py)."""
from .registry import get_class, resolve_model_type


class AutoConfig:
    @classmethod
    def from_pretrained(cls, path: str, **kwargs):
        model_type = resolve_model_type(path)
        config_cls = get_class(model_type, "config")
        return config_cls.from_pretrained(path, **kwargs)
