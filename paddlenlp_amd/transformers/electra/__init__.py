from .configuration import ElectraConfig
from .modeling import (
    ElectraDiscriminator,
    ElectraForSequenceClassification,
    ElectraForTokenClassification,
    ElectraForTotalPretraining,
    ElectraGenerator,
    ElectraModel,
    ElectraPretrainedModel,
)
