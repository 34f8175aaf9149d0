from .prompt_tuning import PromptTuningConfig, PromptTuningModel  # noqa: F401
from .prompt_model import PromptModelForSequenceClassification  # noqa: F401
from .template import ManualTemplate, SoftTemplate, Template  # noqa: F401
from .verbalizer import ManualVerbalizer, SoftVerbalizer  # noqa: F401
