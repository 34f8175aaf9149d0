from .configuration import Qwen2MoeConfig
from .modeling import Qwen2MoeForCausalLM, Qwen2MoeModel, Qwen2MoeSparseMoeBlock
