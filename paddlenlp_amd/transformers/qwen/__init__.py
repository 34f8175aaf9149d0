from .configuration import QWenConfig
from .modeling import QWenForCausalLM, QWenLMHeadModel, QWenModel
