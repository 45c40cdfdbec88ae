"""torchacc_amd — an MI355X-native (CDNA4/gfx950) training-acceleration
framework with the capabilities of AlibabaPAI/torchacc.

One eager PyTorch-ROCm backend: hand-written HIP kernels (MFMA + LDS) for the
hot ops, RCCL over xGMI for every collective, explicit HIP-stream overlap
instead of a tracing compiler. The public surface mirrors the reference
(torchacc/__init__.py): accelerate(), Config, AsyncLoader, amp.GradScaler,
ops.*, dist.*, plus lazy-API shims (sync/mark_step/save/...) that are cheap
no-ops so reference user code runs unchanged.
"""
from typing import Optional

import torch

from .config import (ComputeConfig, Config, DataLoaderConfig, DistConfig,
                     DPConfig, FSDPConfig, MemoryConfig, PPConfig, SPConfig,
                     TPConfig)
from .accelerate import accelerate, broadcast_master_param
from .async_loader import AsyncLoader
from . import amp
from . import dist
from . import ops
from . import utils
from . import llm
from .hf_trainer import accelerate_hf_trainer
from .llm.qwen_patch import patch_qwen_model  # noqa: F401
from .utils.logger import logger

__version__ = "0.1.0"


class GlobalContext:
    """Process-wide context (reference __init__.py:26-37)."""

    def __init__(self):
        self.config: Optional[Config] = None
        self.mesh = None


_global_context = GlobalContext()


def get_global_context() -> GlobalContext:
    return _global_context


# ---------------------------------------------------------------------------
# lazy-API compatibility shims (reference core/__init__.py:12-63). The eager
# backend executes ops immediately; these keep reference user code working.
# ---------------------------------------------------------------------------

def lazy_device() -> torch.device:
    """The compute device (named for reference compatibility)."""
    if torch.cuda.is_available():
        return torch.device("cuda", dist.local_rank())
    return torch.device("cpu")


def sync(wait: bool = False) -> None:
    """Device synchronization point (the reference's mark_step graph cut).
    Eager kernels are already enqueued; wait=True blocks the host."""
    if torch.cuda.is_available() and wait:
        torch.cuda.synchronize()


mark_step = sync


def is_lazy_tensor(t: torch.Tensor) -> bool:
    return False


def save(obj, path, master_only: bool = True) -> None:
    """Rank-aware torch.save (reference ta.save = xm.save)."""
    import torch.distributed as td
    if master_only and td.is_initialized() and td.get_rank() != 0:
        td.barrier()
        return
    torch.save(obj, path)
    if td.is_initialized() and td.get_world_size() > 1:
        td.barrier()


def send_cpu_data_to_device(data, device):
    from .utils.utils import apply_to_tensors
    return apply_to_tensors(lambda t: t.to(device), data)


def fetch_gradients(optimizer):
    grads = []
    for group in optimizer.param_groups:
        for p in group["params"]:
            if p.grad is not None:
                grads.append(p.grad)
    return grads


def mark_dynamic(tensor, dims=None, bounds=None):
    """Bounded-dynamic-shape annotation shim: dynamic shapes are native on
    the eager backend (reference core/dynamic.py:13-46)."""
    return tensor


__all__ = [
    "accelerate", "broadcast_master_param", "AsyncLoader", "Config",
    "ComputeConfig", "MemoryConfig", "DataLoaderConfig", "DistConfig",
    "DPConfig", "TPConfig", "PPConfig", "FSDPConfig", "SPConfig", "amp",
    "dist", "ops", "utils", "lazy_device", "sync", "mark_step",
    "is_lazy_tensor", "save", "send_cpu_data_to_device", "fetch_gradients",
    "mark_dynamic", "get_global_context", "logger",
    "accelerate_hf_trainer", "llm",
]


# import-time side effects (reference __init__.py:135-138): importing the
# package already routes HF transformers' flash-attention entry point (and
# its model-construction availability checks) to the CDNA4 kernels, so
# `attn_implementation="flash_attention_2"` works without the CUDA
# flash_attn package. TORCHACC_PATCH_FA=0 disables (reference patch.py:66).
from .utils import patch as _patch  # noqa: E402

_patch.patch_fa()
_patch.patch_autocast()
