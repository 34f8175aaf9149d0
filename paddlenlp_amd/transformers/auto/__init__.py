from .configuration import AutoConfig  # noqa: F401
from .modeling import AutoModel, AutoModelForCausalLM  # noqa: F401
from .tokenizer import AutoTokenizer  # noqa: F401
