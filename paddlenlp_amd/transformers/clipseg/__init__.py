from .modeling import CLIPSegConfig, CLIPSegForImageSegmentation
