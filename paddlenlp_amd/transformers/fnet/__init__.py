from .modeling import FNetConfig, FNetForMaskedLM, FNetForSequenceClassification, FNetModel
