from .modeling import UNIMOConfig, UNIMOLMHeadModel, UNIMOModel
