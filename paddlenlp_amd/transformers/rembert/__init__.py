from .modeling import (
    RemBertConfig,
    RemBertForMaskedLM,
    RemBertForSequenceClassification,
    RemBertModel,
)
