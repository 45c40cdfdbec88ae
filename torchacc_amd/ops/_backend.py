"""HIP extension loader.

The compiled CDNA4 extension lives IN-TREE as ``torchacc_amd/_C*.so`` (built
by ``python setup.py build_ext --inplace`` / ``__graft_entry__.build()`` with
``PYTORCH_ROCM_ARCH=gfx950``). On a GPU host the ops REQUIRE it — there is no
silent eager fallback on CUDA tensors (a GPU run without the native kernels
is a bug, not a degraded mode). CPU tensors always use the pure-torch
reference math (used by the numerics tests as ground truth).
"""
import importlib

import torch

_C = None
_tried = False


def _load():
    global _C, _tried
    if _tried:
        return _C
    _tried = True
    try:
        _C = importlib.import_module("torchacc_amd._C")
    except ImportError as e:
        _C = None
        _err = e
    return _C


def has_extension() -> bool:
    return _load() is not None


def require_extension():
    ext = _load()
    if ext is None:
        raise RuntimeError(
            "torchacc_amd._C HIP extension is not built. Run "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950) "
            "or `python -c 'import __graft_entry__; __graft_entry__.build()'`."
            " GPU tensors are never silently computed with eager fallbacks.")
    return ext


def dispatch(tensor: torch.Tensor):
    """Return the extension module if ``tensor`` is on GPU (raising if the
    extension is missing), else None (caller uses the CPU reference path)."""
    if tensor.is_cuda:
        return require_extension()
    return None
