from .configuration import BertConfig
from .modeling import (
    BertForMaskedLM,
    BertForMultipleChoice,
    BertForPretraining,
    BertForQuestionAnswering,
    BertForSequenceClassification,
    BertForTokenClassification,
    BertModel,
    BertPretrainedModel,
    BertPretrainingCriterion,
)
