from .argparser import PdArgumentParser  # noqa: F401
from .optimizer import FusedAdamW  # noqa: F401
from .trainer import Trainer  # noqa: F401
from .trainer_callback import (  # noqa: F401
    CallbackHandler,
    DefaultFlowCallback,
    EarlyStoppingCallback,
    ProgressCallback,
    TrainerCallback,
    TrainerControl,
    TrainerState,
)
from .trainer_utils import (  # noqa: F401
    IntervalStrategy,
    ShardingOption,
    get_last_checkpoint,
    get_scheduler,
    set_seed,
    speed_metrics,
)
from .training_args import TrainingArguments  # noqa: F401
from .compression_args import CompressionArguments  # noqa: F401
from .trainer_compress import compress  # noqa: F401
