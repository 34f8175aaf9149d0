from .modeling import (
    MPNetConfig,
    MPNetForMaskedLM,
    MPNetForSequenceClassification,
    MPNetModel,
)
