"""Multi-process test harness (reference tests/utils/distributed.py:14-64).

Single-host multi-process over gloo (CPU) / RCCL (GPU): each test function
runs in ``world_size`` spawned processes with RANK/WORLD_SIZE env + a shared
rendezvous. Exceptions propagate to the parent and fail the test.
"""
import os
import tempfile
import traceback
from typing import Callable

import torch
import torch.multiprocessing as mp


def _worker(rank, world_size, fn, init_file, backend, args, err_q):
    try:
        os.environ["RANK"] = str(rank)
        os.environ["LOCAL_RANK"] = str(rank)
        os.environ["WORLD_SIZE"] = str(world_size)
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.distributed.init_process_group(
            backend=backend, init_method=f"file://{init_file}",
            rank=rank, world_size=world_size)
        try:
            fn(rank, world_size, *args)
        finally:
            torch.distributed.barrier()
            torch.distributed.destroy_process_group()
        err_q.put((rank, None))
    except Exception:
        err_q.put((rank, traceback.format_exc()))


def run_multiprocess(fn: Callable, world_size: int = 2, backend=None,
                     timeout: float = 180.0, args=()):
    """Run ``fn(rank, world_size, *args)`` in ``world_size`` processes."""
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    ctx = mp.get_context("spawn")
    err_q = ctx.Queue()
    with tempfile.NamedTemporaryFile(delete=False) as f:
        init_file = f.name
    os.unlink(init_file)
    procs = [
        ctx.Process(target=_worker,
                    args=(r, world_size, fn, init_file, backend, args, err_q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    errors = []
    for _ in range(world_size):
        rank, err = err_q.get(timeout=timeout)
        if err is not None:
            errors.append(f"rank {rank}:\n{err}")
    for p in procs:
        p.join(timeout=30)
        if p.is_alive():
            p.terminate()
            errors.append("process did not exit")
    assert not errors, "\n".join(errors)
