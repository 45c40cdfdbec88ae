"""Availability probes (reference utils/import_utils.py:6-11)."""
import functools
import importlib


@functools.lru_cache(None)
def is_torch_xla_available() -> bool:
    """Always False: this framework replaces the XLA stack with the eager
    ROCm executor (kept for reference-code compatibility)."""
    return False


@functools.lru_cache(None)
def is_transformers_available() -> bool:
    try:
        importlib.import_module("transformers")
        return True
    except ImportError:
        return False


@functools.lru_cache(None)
def has_hip_extension() -> bool:
    from ..ops._backend import has_extension
    return has_extension()
