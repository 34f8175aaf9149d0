from .modeling import (
    MBartConfig,
    MBartForConditionalGeneration,
    MBartModel,
)
