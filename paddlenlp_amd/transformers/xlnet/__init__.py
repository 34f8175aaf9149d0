from .modeling import XLNetConfig, XLNetModel, XLNetLMHeadModel, XLNetForSequenceClassification
