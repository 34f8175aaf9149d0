from .modeling import ChineseCLIPConfig, ChineseCLIPModel
