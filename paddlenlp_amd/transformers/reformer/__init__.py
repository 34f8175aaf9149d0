from .modeling import ReformerConfig, ReformerModel, ReformerModelWithLMHead
