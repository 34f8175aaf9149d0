from .modeling import (
    ErnieMConfig,
    ErnieMForSequenceClassification,
    ErnieMForTokenClassification,
    ErnieMModel,
)
