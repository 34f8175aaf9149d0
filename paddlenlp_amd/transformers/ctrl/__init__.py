from .modeling import CTRLConfig, CTRLModel, CTRLLMHeadModel, CTRLForSequenceClassification
