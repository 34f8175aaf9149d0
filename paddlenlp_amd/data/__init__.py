from .data_collator import (  # noqa: F401
    DataCollatorForLanguageModeling,
    DataCollatorForSeq2Seq,
    DataCollatorForTokenClassification,
    DataCollatorForWholeWordMask,
    DataCollatorWithPadding,
    default_data_collator,
)
from .sampler import DistributedBatchSampler  # noqa: F401
from .collate import (  # noqa: F401
    Dict,
    JiebaLikeTokenizer,
    Pad,
    Stack,
    Tuple,
    Vocab,
)
