from .tokenizer import BertJapaneseTokenizer
