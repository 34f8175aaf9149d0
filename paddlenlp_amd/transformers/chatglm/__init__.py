from .modeling import ChatGLMConfig, ChatGLMForCausalLM, ChatGLMModel
