from .configuration import MixtralConfig  # noqa: F401
from .modeling import (  # noqa: F401
    MixtralForCausalLM,
    MixtralModel,
    MixtralSparseMoeBlock,
    load_balancing_loss_func,
)
