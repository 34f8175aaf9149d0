from .configuration import LlamaConfig  # noqa: F401
from .modeling import (  # noqa: F401
    LlamaAttention,
    LlamaDecoderLayer,
    LlamaForCausalLM,
    LlamaMLP,
    LlamaModel,
    LlamaPretrainingCriterion,
    LlamaRMSNorm,
)
