from .modeling import (
    NeZhaConfig,
    NeZhaForQuestionAnswering,
    NeZhaForSequenceClassification,
    NeZhaForTokenClassification,
    NeZhaModel,
)
