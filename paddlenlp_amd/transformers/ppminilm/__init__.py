from .modeling import (
    PPMiniLMConfig,
    PPMiniLMForQuestionAnswering,
    PPMiniLMForSequenceClassification,
    PPMiniLMModel,
)
