from .configuration import BloomConfig
from .modeling import BloomForCausalLM, BloomModel
