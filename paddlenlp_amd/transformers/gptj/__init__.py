from .configuration import GPTJConfig
from .modeling import GPTJForCausalLM, GPTJModel
