from .server import SimpleServer  # noqa: F401
