from .modeling import (
    TinyBertConfig,
    TinyBertForPretraining,
    TinyBertForQuestionAnswering,
    TinyBertForSequenceClassification,
    TinyBertModel,
)
