from .modeling import ErnieCtmConfig, ErnieCtmModel, ErnieCtmWordtagModel
