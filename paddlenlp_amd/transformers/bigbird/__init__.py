from .modeling import BigBirdConfig, BigBirdModel, BigBirdForSequenceClassification, BigBirdForMaskedLM
