from .configuration import MambaConfig
from .modeling import MambaForCausalLM, MambaMixer, MambaModel
