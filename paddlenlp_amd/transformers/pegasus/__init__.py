from .modeling import (
    PegasusConfig,
    PegasusForConditionalGeneration,
    PegasusModel,
)
