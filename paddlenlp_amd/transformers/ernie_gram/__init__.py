from .modeling import (
    ErnieGramConfig,
    ErnieGramForQuestionAnswering,
    ErnieGramForSequenceClassification,
    ErnieGramForTokenClassification,
    ErnieGramModel,
)
