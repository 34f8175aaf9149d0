from .modeling import BitConfig, BitModel, BitForImageClassification
