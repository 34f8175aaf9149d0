from .configuration import AlbertConfig
from .modeling import (
    AlbertForMaskedLM,
    AlbertForSequenceClassification,
    AlbertModel,
)
