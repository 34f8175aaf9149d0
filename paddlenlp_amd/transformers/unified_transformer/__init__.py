from .modeling import (
    UnifiedTransformerConfig,
    UnifiedTransformerLMHeadModel,
    UnifiedTransformerModel,
)
