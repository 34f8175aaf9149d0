from .modeling import YuanConfig, YuanForCausalLM, YuanModel
