from .configuration import T5Config
from .modeling import T5EncoderModel, T5ForConditionalGeneration, T5Model
