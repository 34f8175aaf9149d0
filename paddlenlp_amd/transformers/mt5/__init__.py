from .modeling import MT5Config, MT5EncoderModel, MT5ForConditionalGeneration, MT5Model
