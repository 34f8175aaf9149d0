from .modeling import (
    BlenderbotConfig,
    BlenderbotForConditionalGeneration,
    BlenderbotModel,
)
