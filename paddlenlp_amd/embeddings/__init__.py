from .token_embedding import TokenEmbedding  # noqa: F401
