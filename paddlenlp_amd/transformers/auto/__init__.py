from .configuration import AutoConfig  # noqa: F401
from .modeling import (  # noqa: F401
    AutoModel,
    AutoModelForCausalLM,
    AutoModelForConditionalGeneration,
    AutoModelForMaskedLM,
    AutoModelForMultipleChoice,
    AutoModelForSeq2SeqLM,
    AutoModelForQuestionAnswering,
    AutoModelForSequenceClassification,
    AutoModelForTokenClassification,
)
from .tokenizer import AutoTokenizer  # noqa: F401
