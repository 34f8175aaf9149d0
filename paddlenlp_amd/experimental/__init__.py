from .block_manager import BlockManager  # noqa: F401
from .fused_transformer import (  # noqa: F401
    FusedMultiTransformer,
    FusedMultiTransformerConfig,
    GraphDecodeRunner,
)
