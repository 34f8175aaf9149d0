from .configuration import DeepseekV2Config
from .modeling import DeepseekV2ForCausalLM, DeepseekV2Model
