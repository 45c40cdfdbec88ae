"""Communication backend: RCCL-over-xGMI stream management + bucketing.

The reference needed a custom c10d backend ('lazy', dist/backend.py:34-421)
to record collectives into the XLA graph and to emulate send/recv as masked
all-reduces. On the eager MI355X build none of that is needed: RCCL (torch
backend "nccl" on ROCm) has native send/recv and every collective, and overlap
is engineered explicitly with HIP side-streams + events instead of XLA's
latency-hiding scheduler. This module provides:

- :class:`CommStreams`: named side streams (gradient reduction, FSDP param
  all-gather prefetch, d2h/h2d offload) with event-based hand-off helpers.
- :class:`BucketedAllReduce`: coalesces many small grads into flat buckets
  sized for xGMI (7 links x ~153 GB/s per GPU -> large buckets amortize
  per-collective latency; default 64 MiB) and overlaps reduction with
  backward compute.
"""
from typing import Dict, List, Optional

import torch
import torch.distributed as dist

_STREAMS: Dict[str, torch.cuda.Stream] = {}


def _cuda_ok() -> bool:
    return torch.cuda.is_available()


def get_comm_stream(name: str = "comm") -> Optional[torch.cuda.Stream]:
    """A named HIP side stream for communication (None on CPU)."""
    if not _cuda_ok():
        return None
    s = _STREAMS.get(name)
    if s is None:
        s = torch.cuda.Stream()
        _STREAMS[name] = s
    return s


class CommStreams:
    """Event-based compute<->comm hand-off.

    Usage on the comm side::

        with comm.on("reduce"):        # waits for current compute stream
            dist.all_reduce(buf, group=g, async_op=False)
    """

    def __init__(self):
        self._events: List[torch.cuda.Event] = []

    class _Ctx:
        def __init__(self, stream):
            self.stream = stream

        def __enter__(self):
            if self.stream is not None:
                self.stream.wait_stream(torch.cuda.current_stream())
                self._guard = torch.cuda.stream(self.stream)
                self._guard.__enter__()
            return self

        def __exit__(self, *a):
            if self.stream is not None:
                self._guard.__exit__(*a)

    def on(self, name: str) -> "_Ctx":
        return self._Ctx(get_comm_stream(name))

    @staticmethod
    def join(name: str):
        """Make the current compute stream wait for the named comm stream."""
        s = get_comm_stream(name)
        if s is not None:
            torch.cuda.current_stream().wait_stream(s)


class Bucket:
    __slots__ = ("flat", "grads", "numel", "ready", "work")

    def __init__(self, capacity: int, dtype, device):
        self.flat = torch.zeros(capacity, dtype=dtype, device=device)
        self.grads: List[torch.Tensor] = []
        self.numel = 0
        self.ready = 0
        self.work = None


class BucketedAllReduce:
    """Coalesced, overlapped gradient all-reduce for DP.

    Gradients are copied into a flat bucket as their backward hooks fire; when
    a bucket fills, an async all-reduce of the flat buffer launches on a side
    stream. ``finalize()`` waits for all buckets, scales by 1/N and copies the
    reduced values back into the parameter .grad tensors.

    Matches the semantics of the reference's lazy DP (coalesced all-reduce of
    all grads then grad /= dp_size, dist/dp.py:40-71) but overlaps per-bucket
    with backward.
    """

    def __init__(self, group, bucket_mb: int = 64):
        self.group = group
        self.bucket_bytes = bucket_mb * 1024 * 1024
        self._pending: List[Bucket] = []
        self._current: Dict[torch.dtype, Bucket] = {}

    def _flush(self, bucket: Bucket):
        if bucket.numel == 0:
            return
        view = bucket.flat[:bucket.numel]
        if _cuda_ok():
            s = get_comm_stream("reduce")
            s.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(s):
                bucket.work = dist.all_reduce(
                    view, group=self.group, async_op=True)
        else:
            bucket.work = dist.all_reduce(
                view, group=self.group, async_op=True)
        self._pending.append(bucket)

    def add(self, grad: torch.Tensor):
        dtype = grad.dtype
        cap = max(self.bucket_bytes // grad.element_size(), grad.numel())
        b = self._current.get(dtype)
        if b is None or b.numel + grad.numel() > b.flat.numel():
            if b is not None:
                self._flush(b)
            b = Bucket(cap, dtype, grad.device)
            self._current[dtype] = b
        n = grad.numel()
        b.flat[b.numel:b.numel + n].copy_(grad.reshape(-1))
        b.grads.append(grad)
        b.numel += n

    def finalize(self, world: int):
        for b in list(self._current.values()):
            self._flush(b)
        self._current.clear()
        if _cuda_ok():
            torch.cuda.current_stream().wait_stream(get_comm_stream("reduce"))
        for b in self._pending:
            if b.work is not None:
                b.work.wait()
            off = 0
            scale = 1.0 / world
            for g in b.grads:
                n = g.numel()
                g.reshape(-1).copy_(b.flat[off:off + n]).mul_(scale)
                off += n
        self._pending.clear()


def all_reduce_coalesced(tensors: List[torch.Tensor],
                         group=None,
                         op=dist.ReduceOp.SUM):
    """One flat all-reduce over many tensors (reference backend.py:200)."""
    if not tensors:
        return
    flat = torch.cat([t.reshape(-1) for t in tensors])
    dist.all_reduce(flat, op=op, group=group)
    off = 0
    for t in tensors:
        n = t.numel()
        t.copy_(flat[off:off + n].view_as(t))
        off += n
