from .rdrop import RDropLoss  # noqa: F401
