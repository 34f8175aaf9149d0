from .configuration import BlipConfig, BlipTextConfig, BlipVisionConfig
from .modeling import (
    BlipForConditionalGeneration,
    BlipForImageTextRetrieval,
    BlipModel,
)
