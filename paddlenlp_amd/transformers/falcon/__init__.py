from .configuration import FalconConfig
from .modeling import FalconForCausalLM, FalconModel
