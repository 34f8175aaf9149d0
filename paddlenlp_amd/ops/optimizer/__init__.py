from .adamwdl import AdamWDL  # noqa: F401
