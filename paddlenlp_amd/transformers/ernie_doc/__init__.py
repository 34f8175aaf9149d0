from .modeling import ErnieDocConfig, ErnieDocModel, ErnieDocForSequenceClassification
