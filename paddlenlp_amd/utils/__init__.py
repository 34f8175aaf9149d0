from .log import logger  # noqa: F401
