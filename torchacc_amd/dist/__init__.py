"""Distributed init + helpers (reference torchacc/dist/__init__.py:33-116).

One process per GPU; ``torch.distributed`` backend "nccl" is RCCL on ROCm
(xGMI intra-node), "gloo" on CPU-only hosts (used by the multi-process unit
tests).
"""
import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist

from ..utils.logger import logger
from .backend import (BucketedAllReduce, CommStreams, all_reduce_coalesced,
                      get_comm_stream)
from .mesh import Mesh, ProcessTopology

__all__ = [
    "world_size", "rank", "local_rank", "init_process_group",
    "init_comm_context", "init_nccl_context", "rendezvous", "Mesh",
    "ProcessTopology", "BucketedAllReduce", "CommStreams",
    "all_reduce_coalesced", "get_comm_stream", "DataParallel",
    "FullyShardedDataParallel", "DistributedParallel", "PipelineParallel",
]


def world_size() -> int:
    if dist.is_initialized():
        return dist.get_world_size()
    return int(os.environ.get("WORLD_SIZE", 1))


def rank() -> int:
    if dist.is_initialized():
        return dist.get_rank()
    return int(os.environ.get("RANK", 0))


def local_rank() -> int:
    return int(os.environ.get("LOCAL_RANK", 0))


def _default_backend() -> str:
    return "nccl" if torch.cuda.is_available() else "gloo"


def init_process_group(config=None, backend: Optional[str] = None,
                       timeout_s: int = 1800):
    """Idempotent process-group init (reference dist/__init__.py:45-55)."""
    if dist.is_initialized():
        return
    if world_size() == 1 and "MASTER_ADDR" not in os.environ:
        # degenerate single-process case: still init so collectives are no-ops
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29577")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
    backend = backend or _default_backend()
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank())
    dist.init_process_group(
        backend=backend,
        timeout=datetime.timedelta(seconds=timeout_s))
    logger.info("init_process_group backend=%s rank=%d/%d", backend, rank(),
                world_size())


def init_comm_context(config) -> None:
    """Warm up communication cliques so the first real collective in the hot
    loop doesn't pay RCCL communicator setup (reference init_nccl_context,
    dist/__init__.py:58-98). On RCCL the warm-up is a tiny all-reduce per
    group plus a PP-neighbor send/recv ring.
    """
    if not dist.is_initialized() or world_size() == 1:
        return
    mesh = config.get_mesh()
    device = torch.device("cuda", local_rank()) \
        if torch.cuda.is_available() else torch.device("cpu")
    one = torch.ones(1, device=device)
    for group in (mesh.get_dp_proc_group(), mesh.get_fsdp_proc_group(),
                  mesh.get_tp_proc_group()):
        if group is not None:
            dist.all_reduce(one.clone(), group=group)
    if mesh.get_pp_num() > 1:
        # neighbor send/recv warm-up (wraparound ring), even/odd ordering
        stage = mesh.get_stage_id()
        nstages = mesh.get_pp_num()
        nxt = mesh.stage_to_global((stage + 1) % nstages)
        prv = mesh.stage_to_global((stage - 1) % nstages)
        buf = torch.zeros(1, device=device)
        if stage % 2 == 0:
            dist.send(one, nxt)
            dist.recv(buf, prv)
        else:
            dist.recv(buf, prv)
            dist.send(one, nxt)
    if torch.cuda.is_available():
        torch.cuda.synchronize()


# Reference-compatible alias (the reference named it after NCCL).
init_nccl_context = init_comm_context


def rendezvous(tag: str = "") -> None:
    """Barrier across all ranks (reference dist/__init__.py:101-116)."""
    if dist.is_initialized() and world_size() > 1:
        dist.barrier()


# strategy wrappers (imported late to avoid cycles)
from .parallel_module import ParallelModule  # noqa: E402,F401
from .dp import DataParallel  # noqa: E402
from .fsdp import FullyShardedDataParallel  # noqa: E402
from .spmd_fsdp import SpmdFullyShardedDataParallel  # noqa: E402
from .distributed_parallel import DistributedParallel  # noqa: E402
from .pp.pipeline import PipelineParallel  # noqa: E402
from . import tp  # noqa: E402
from . import state_dict_utils  # noqa: E402

# reference-compatible backend-name constants: on the eager MI355X build
# both map to RCCL ("nccl" IS RCCL on ROCm); the reference distinguished
# its registered 'lazy' c10d backend from eager NCCL (dist/__init__.py:28-29)
BACKEND_NAME = "nccl"
EAGER_BACKEND_NAME = "nccl"
