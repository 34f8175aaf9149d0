from .configuration import GPTConfig  # noqa: F401
from .modeling import GPTForCausalLM, GPTModel, GPTPretrainingCriterion  # noqa: F401
