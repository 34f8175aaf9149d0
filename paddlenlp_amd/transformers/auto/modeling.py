"""AutoModel / AutoModelForCausalLM (reference: auto/modeling.py:405)."""
from .registry import get_class, resolve_model_type


class _AutoBase:
    _kind = "base"

    @classmethod
    def from_pretrained(cls, path: str, **kwargs):
        model_type = resolve_model_type(path)
        model_cls = get_class(model_type, cls._kind)
        return model_cls.from_pretrained(path, **kwargs)

    @classmethod
    def from_config(cls, config, **kwargs):
        model_cls = get_class(config.model_type, cls._kind)
        return model_cls.from_config(config, **kwargs)


class AutoModel(_AutoBase):
    _kind = "base"


class AutoModelForCausalLM(_AutoBase):
    _kind = "causal_lm"


class AutoModelForSeq2SeqLM(_AutoBase):
    _kind = "seq2seq_lm"


class AutoModelForConditionalGeneration(AutoModelForSeq2SeqLM):
    """Reference's name for the seq2seq auto class (auto/modeling.py)."""


class AutoModelForSequenceClassification(_AutoBase):
    _kind = "sequence_classification"


class AutoModelForTokenClassification(_AutoBase):
    _kind = "token_classification"


class AutoModelForQuestionAnswering(_AutoBase):
    _kind = "question_answering"


class AutoModelForMultipleChoice(_AutoBase):
    _kind = "multiple_choice"


class AutoModelForMaskedLM(_AutoBase):
    _kind = "masked_lm"
