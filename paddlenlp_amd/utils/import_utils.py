"""Import availability helpers (reference: paddlenlp/utils/import_utils.py)."""
from __future__ import annotations

import importlib
import importlib.util
from functools import lru_cache


@lru_cache(maxsize=None)
def is_package_available(name: str) -> bool:
    return importlib.util.find_spec(name) is not None


def is_transformers_available() -> bool:
    return is_package_available("transformers")


def is_datasets_available() -> bool:
    return is_package_available("datasets")


def is_sentencepiece_available() -> bool:
    return is_package_available("sentencepiece")


def import_module(name: str):
    """Import a dotted module path, returning None when unavailable."""
    try:
        return importlib.import_module(name)
    except ImportError:
        return None
