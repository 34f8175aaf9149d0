from .modeling import (
    SkepConfig,
    SkepCrfForTokenClassification,
    SkepForSequenceClassification,
    SkepForTokenClassification,
    SkepModel,
)
