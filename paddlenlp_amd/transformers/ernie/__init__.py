from .configuration import ErnieConfig
from .modeling import (
    UIE,
    ErnieForMaskedLM,
    ErnieForPretraining,
    ErnieForQuestionAnswering,
    ErnieForSequenceClassification,
    ErnieForTokenClassification,
    ErnieModel,
    ErniePretrainedModel,
    ErniePretrainingCriterion,
)
