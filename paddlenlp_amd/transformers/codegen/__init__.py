from .modeling import CodeGenConfig, CodeGenForCausalLM, CodeGenModel
