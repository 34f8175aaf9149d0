from .modeling import TransformerConfig, TransformerModel
