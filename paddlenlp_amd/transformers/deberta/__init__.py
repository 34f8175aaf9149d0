from .modeling import (
    DebertaConfig,
    DebertaForMaskedLM,
    DebertaForSequenceClassification,
    DebertaModel,
)
