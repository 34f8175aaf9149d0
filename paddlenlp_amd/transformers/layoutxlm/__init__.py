"""LayoutXLM = multilingual LayoutLMv2 (reference layoutxlm/ shares the
layoutlmv2 architecture with an XLM-R vocabulary)."""
from ..layoutlmv2.modeling import (
    LayoutLMv2Config,
    LayoutLMv2ForTokenClassification,
    LayoutLMv2Model,
)


class LayoutXLMConfig(LayoutLMv2Config):
    model_type = "layoutxlm"

    def __init__(self, vocab_size=250002, **kwargs):
        super().__init__(vocab_size=vocab_size, **kwargs)


class LayoutXLMModel(LayoutLMv2Model):
    config_class = LayoutXLMConfig
    base_model_prefix = "layoutxlm"


class LayoutXLMForTokenClassification(LayoutLMv2ForTokenClassification):
    config_class = LayoutXLMConfig
    base_model_prefix = "layoutxlm"
