from .metrics import BLEU, AccuracyAndF1, ChunkEvaluator, Perplexity, Rouge1, RougeL  # noqa: F401
