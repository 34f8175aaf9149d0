from .text_classification import AutoTrainerForTextClassification  # noqa: F401
