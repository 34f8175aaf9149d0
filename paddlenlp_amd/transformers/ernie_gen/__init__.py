from .modeling import ErnieGenConfig, ErnieGenModel, ErnieGenForGeneration
