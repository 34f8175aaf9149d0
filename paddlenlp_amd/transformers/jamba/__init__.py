from .configuration import JambaConfig
from .modeling import JambaForCausalLM, JambaModel
