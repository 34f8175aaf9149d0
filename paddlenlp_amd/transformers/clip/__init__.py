from .configuration import CLIPConfig, CLIPTextConfig, CLIPVisionConfig
from .modeling import CLIPModel, CLIPTextModel, CLIPVisionModel
