from .configuration import ErnieConfig
from .modeling import (
    UIE,
    UTC,
    ErnieForMaskedLM,
    ErnieForPretraining,
    ErnieForQuestionAnswering,
    ErnieForSequenceClassification,
    ErnieForTokenClassification,
    ErnieModel,
    ErniePretrainedModel,
    ErniePretrainingCriterion,
)
