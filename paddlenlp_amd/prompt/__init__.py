from .prompt_tuning import PromptTuningConfig, PromptTuningModel  # noqa: F401
