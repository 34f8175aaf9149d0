from .configuration import GemmaConfig
from .modeling import GemmaForCausalLM, GemmaModel
