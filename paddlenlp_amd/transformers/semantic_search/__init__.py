from .modeling import ErnieDualEncoder, ErnieCrossEncoder
