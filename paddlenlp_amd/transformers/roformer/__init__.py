from .modeling import (
    RoFormerConfig,
    RoFormerForMaskedLM,
    RoFormerForSequenceClassification,
    RoFormerModel,
)
