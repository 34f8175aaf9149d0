from .modeling import (
    MegatronBertConfig,
    MegatronBertForQuestionAnswering,
    MegatronBertForSequenceClassification,
    MegatronBertModel,
)
