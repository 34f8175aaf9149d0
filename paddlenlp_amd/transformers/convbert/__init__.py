from .modeling import ConvBertConfig, ConvBertModel, ConvBertForSequenceClassification, ConvBertForMaskedLM
