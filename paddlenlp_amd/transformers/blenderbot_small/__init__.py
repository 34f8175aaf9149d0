from .modeling import (
    BlenderbotSmallConfig,
    BlenderbotSmallForConditionalGeneration,
    BlenderbotSmallModel,
)
