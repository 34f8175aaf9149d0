from .modeling import GAUAlphaConfig, GAUAlphaModel, GAUAlphaForSequenceClassification
