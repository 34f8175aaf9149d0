"""UNIMO family (reference: paddlenlp/transformers/unimo/modeling.py).

Unified-modal text tower: same PREFIX-LM architecture as
UnifiedTransformer (pre-LN stack, word+position+token_type embeddings,
seq2seq attention mask, tied LM head) with unimo config defaults.
"""
from __future__ import annotations

from ..unified_transformer.modeling import (
    UnifiedTransformerConfig,
    UnifiedTransformerLMHeadModel,
    UnifiedTransformerModel,
)

__all__ = ["UNIMOConfig", "UNIMOModel", "UNIMOLMHeadModel"]


class UNIMOConfig(UnifiedTransformerConfig):
    model_type = "unimo"

    def __init__(self, vocab_size=18000, **kwargs):
        kwargs.setdefault("type_vocab_size", 4)
        super().__init__(vocab_size=vocab_size, **kwargs)


class UNIMOModel(UnifiedTransformerModel):
    config_class = UNIMOConfig
    base_model_prefix = "unimo"


class UNIMOLMHeadModel(UnifiedTransformerLMHeadModel):
    config_class = UNIMOConfig
    base_model_prefix = "unimo"

    def __init__(self, config):
        super().__init__(config)
        # keep the reference attribute name
        self.unimo = self.unified_transformer
