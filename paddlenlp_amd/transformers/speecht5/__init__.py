from .modeling import SpeechT5Config, SpeechT5Model, SpeechT5ForSpeechToText, SpeechT5ForTextToSpeech
