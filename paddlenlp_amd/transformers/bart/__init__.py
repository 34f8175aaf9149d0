from .configuration import BartConfig
from .modeling import BartDecoder, BartEncoder, BartForConditionalGeneration, BartModel
