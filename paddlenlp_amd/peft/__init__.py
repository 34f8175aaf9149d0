from .lora import LoRAConfig, LoRAModel  # noqa: F401
from .prefix import PrefixConfig, PrefixModelForCausalLM  # noqa: F401
from .vera import VeRAConfig, VeRAModel  # noqa: F401
