"""AutoTokenizer (reference: paddlenlp/transformers/auto/tokenizer.py)."""
from ..tokenizer_utils import PretrainedTokenizer


class AutoTokenizer:
    @classmethod
    def from_pretrained(cls, path: str, **kwargs):
        return PretrainedTokenizer.from_pretrained(path, **kwargs)
