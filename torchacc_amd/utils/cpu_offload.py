"""Activation CPU offload via saved-tensor pack/unpack hooks.

Reimplements the reference's utils/cpu_offload.py:18-605 (itself
TransformerEngine-derived) natively for HIP streams:

- saved activations are offloaded device->host on a dedicated "offload_d2h"
  HIP stream into pinned host buffers, grouped per layer-group commit;
- during backward, groups are prefetched host->device ``num_prefetch_group``
  groups ahead on "offload_h2d";
- tensors are deduped by (data_ptr, shape, stride, dtype) within a group
  (identical views move once; distinct views of one storage move each);
- user API: ``get_cpu_offload_context(num_offload_layers,
  num_prefetch_layers, ...)`` returning (context, sync_fn) — wrap each
  layer's forward in the context and call sync_fn between layer groups.

On CPU-only hosts the machinery degrades to a no-op (tensors stay put) so
the same code paths are unit-testable without a GPU.
"""
from typing import Dict, Optional

import torch

from ..dist.backend import get_comm_stream
from .logger import logger


class _OffloadGroup:
    def __init__(self, idx: int):
        self.idx = idx
        self.packed: Dict[tuple, tuple] = {}  # view key -> (host, meta)
        self.d2h_event: Optional[torch.cuda.Event] = None
        self.h2d_event: Optional[torch.cuda.Event] = None
        self.device_cache: Dict[tuple, torch.Tensor] = {}


class AsyncDoubleBufferGroupOffloadHandler:
    """Bulk offload per layer group with double-buffered streams
    (reference :310-519)."""

    def __init__(self, num_offload_group: int, num_prefetch_group: int = 1,
                 num_sync_group: int = 0):
        self.num_offload_group = num_offload_group
        self.num_prefetch_group = max(1, num_prefetch_group)
        # first N groups offload synchronously (reference
        # num_offload_sync_layers semantics, cpu_offload.py:521-605):
        # their activations are guaranteed resident on host before the
        # forward proceeds — bounding device memory exactly at the cost of
        # overlap
        self.num_sync_group = num_sync_group
        self.groups: Dict[int, _OffloadGroup] = {}
        self.current_group = 0
        self._on_gpu = torch.cuda.is_available()

    # ---- forward side ---------------------------------------------------

    def offload_enabled(self, group_idx: int) -> bool:
        return group_idx < self.num_offload_group

    def tensor_push(self, tensor: torch.Tensor) -> tuple:
        gid = self.current_group
        if not self._on_gpu or not self.offload_enabled(gid) or \
                not tensor.is_cuda or tensor.numel() < 1024:
            return ("keep", tensor)
        if gid < self.num_sync_group:
            host = tensor.detach().cpu()
            return ("sync_offloaded", host, tensor.device, tensor.dtype)
        grp = self.groups.setdefault(gid, _OffloadGroup(gid))
        # dedupe key includes the VIEW's geometry: two distinct views of one
        # storage (same data_ptr, different shape/stride) must offload
        # separately or the pop returns the wrong tensor (the reference
        # tracks per-tensor metadata, cpu_offload.py:310-519)
        key = (tensor.data_ptr(), tuple(tensor.shape),
               tuple(tensor.stride()), tensor.dtype)
        if key not in grp.packed:
            host = torch.empty(tensor.shape, dtype=tensor.dtype,
                               device="cpu", pin_memory=True)
            s = get_comm_stream("offload_d2h")
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                host.copy_(tensor, non_blocking=True)
                tensor.record_stream(s)
            grp.packed[key] = (host, tensor.device)
            logger.debug("offload push group %d: %s", gid,
                         tuple(tensor.shape))
        return ("offloaded", gid, key, tensor.shape, tensor.dtype)

    def commit_group(self):
        """Mark the end of a layer group's forward; records the d2h event
        (reference GroupCommitFunction :143-164)."""
        gid = self.current_group
        if self._on_gpu and gid in self.groups:
            ev = torch.cuda.Event()
            ev.record(get_comm_stream("offload_d2h"))
            self.groups[gid].d2h_event = ev
        self.current_group += 1

    # ---- backward side --------------------------------------------------

    def start_backward(self):
        self.current_group -= 1
        self._prefetch_upto = self.current_group

    def _prefetch_group(self, gid: int):
        grp = self.groups.get(gid)
        if grp is None or grp.h2d_event is not None:
            return
        s = get_comm_stream("offload_h2d")
        if grp.d2h_event is not None:
            s.wait_event(grp.d2h_event)
        with torch.cuda.stream(s):
            for key, (host, device) in grp.packed.items():
                dev = torch.empty(host.shape, dtype=host.dtype,
                                  device=device)
                dev.copy_(host, non_blocking=True)
                grp.device_cache[key] = dev
        ev = torch.cuda.Event()
        ev.record(s)
        grp.h2d_event = ev

    def tensor_pop(self, packed) -> torch.Tensor:
        if packed[0] == "keep":
            return packed[1]
        if packed[0] == "sync_offloaded":
            return packed[1].to(packed[2])
        _, gid, key, shape, dtype = packed
        grp = self.groups[gid]
        # prefetch this group + the next ones toward group 0
        for g in range(gid, max(-1, gid - 1 - self.num_prefetch_group), -1):
            self._prefetch_group(g)
        if grp.h2d_event is not None:
            torch.cuda.current_stream().wait_event(grp.h2d_event)
        return grp.device_cache[key]

    def reset(self):
        self.groups.clear()
        self.current_group = 0


class SynchronizedGroupOffloadHandler(AsyncDoubleBufferGroupOffloadHandler):
    """Blocking variant (reference :167-252): d2h/h2d on the default stream
    with host syncs — simpler, slower, for debugging."""

    def tensor_push(self, tensor):
        gid = self.current_group
        if not self._on_gpu or not self.offload_enabled(gid) or \
                not tensor.is_cuda:
            return ("keep", tensor)
        host = tensor.detach().cpu()
        return ("sync_offloaded", host, tensor.device, tensor.dtype)

    def tensor_pop(self, packed):
        if packed[0] == "keep":
            return packed[1]
        _, host, device, dtype = packed
        return host.to(device)


class _OffloadContext:
    def __init__(self, handler):
        self.handler = handler
        self._hooks_ctx = None

    def __enter__(self):
        h = self.handler

        def pack(t):
            return h.tensor_push(t)

        def unpack(p):
            return h.tensor_pop(p)

        self._hooks_ctx = torch.autograd.graph.saved_tensors_hooks(
            pack, unpack)
        self._hooks_ctx.__enter__()
        return self

    def __exit__(self, *a):
        self._hooks_ctx.__exit__(*a)


def get_cpu_offload_context(num_offload_layers: int = 1,
                            num_prefetch_layers: int = 1,
                            num_offload_sync_layers: int = 0,
                            synchronous: bool = False):
    """Returns (context, group_commit_fn).

    Usage (reference :521-605)::

        ctx, commit = get_cpu_offload_context(num_layers - 1)
        for layer in layers:
            with ctx:
                x = layer(x)
            x = commit(x)    # marks the layer-group boundary
    """
    if synchronous:
        handler = SynchronizedGroupOffloadHandler(
            num_offload_layers, num_prefetch_layers)
    else:
        handler = AsyncDoubleBufferGroupOffloadHandler(
            num_offload_layers, num_prefetch_layers,
            num_sync_group=num_offload_sync_layers)

    class _Commit(torch.autograd.Function):
        @staticmethod
        def forward(ctx, x):
            handler.commit_group()
            return x

        @staticmethod
        def backward(ctx, g):
            handler.current_group -= 1
            return g

    def group_commit(x):
        if isinstance(x, torch.Tensor) and x.requires_grad:
            return _Commit.apply(x)
        handler.commit_group()
        return x

    return _OffloadContext(handler), group_commit
