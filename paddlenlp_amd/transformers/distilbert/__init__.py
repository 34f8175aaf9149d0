from .modeling import (
    DistilBertConfig,
    DistilBertForMaskedLM,
    DistilBertForSequenceClassification,
    DistilBertModel,
)
