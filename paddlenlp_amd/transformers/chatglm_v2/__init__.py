from .configuration import ChatGLMv2Config
from .modeling import ChatGLMv2ForCausalLM, ChatGLMv2Model
