from .configuration import Qwen2Config  # noqa: F401
from .modeling import Qwen2ForCausalLM, Qwen2Model  # noqa: F401
