from .configuration import ErnieLayoutConfig
from .modeling import (
    ErnieLayoutForQuestionAnswering,
    ErnieLayoutForTokenClassification,
    ErnieLayoutModel,
)
