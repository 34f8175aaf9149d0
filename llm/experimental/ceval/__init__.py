from .eval import evaluate_mcq, load_ceval_split  # noqa: F401
