from .modeling import (
    XLMConfig,
    XLMForSequenceClassification,
    XLMModel,
    XLMWithLMHeadModel,
)
