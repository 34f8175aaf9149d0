from .functional import (  # noqa: F401
    build_rope_cache,
    cross_entropy,
    extension_available,
    flash_attention,
    fused_adamw,
    fused_rope,
    rms_norm,
    swiglu,
)
from . import reference  # noqa: F401
