from .configuration import OPTConfig
from .modeling import OPTForCausalLM, OPTModel
