from .configuration import MistralConfig  # noqa: F401
from .modeling import MistralForCausalLM, MistralModel  # noqa: F401
