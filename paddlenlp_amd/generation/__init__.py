from .configuration_utils import GenerationConfig  # noqa: F401
from .logits_process import (  # noqa: F401
    LogitsProcessorList,
    RepetitionPenaltyLogitsProcessor,
    TemperatureLogitsWarper,
    TopKLogitsWarper,
    TopPLogitsWarper,
)
from .utils import GenerationMixin  # noqa: F401
from .stopping_criteria import (  # noqa: F401
    MaxLengthCriteria,
    MaxNewTokensCriteria,
    MaxTimeCriteria,
    StoppingCriteria,
    StoppingCriteriaList,
    StopStringsCriteria,
)
