from .word import (  # noqa: F401
    WordDelete,
    WordInsert,
    WordSubstitute,
    WordSwap,
)
from .char import (  # noqa: F401
    CharDelete,
    CharInsert,
    CharSubstitute,
    CharSwap,
)
