from .modeling import GLMConfig, GLMModel, GLMForConditionalGeneration
