from .modeling import ErnieViLConfig, ErnieViLModel
