"""PP point-to-point communication (reference dist/pp/p2p.py:7-37).

RCCL has native send/recv on MI355X (no masked-all-reduce emulation).
Activation messages carry a FIXED-SIZE int64 meta header (tensor count +
per-tensor dims/dtype/shape) followed by the payloads — the reference's
tensor-meta handshake (executor.py:475-570). Gradient messages need no
meta (the receiver allocated buffers from the shapes it sent forward).

``send_acts_recv_grads`` fuses the steady-state 1F1B pair — send
activations downstream while receiving gradients from downstream — into one
``batch_isend_irecv`` so neither side's blocking order can deadlock
(the reference handled this with its even/odd op ordering,
utils.py:403-408; on a chain the fused batch is the standard solution).
"""
from typing import List

import torch
import torch.distributed as dist

_DTYPES = [
    torch.float32, torch.float16, torch.bfloat16, torch.int64, torch.int32,
    torch.int16, torch.int8, torch.uint8, torch.bool, torch.float64,
]
_MAX_DIMS = 8
_MAX_TENSORS = 24
_META_LEN = 1 + _MAX_TENSORS * (2 + _MAX_DIMS)


def _meta_of(tensors: List[torch.Tensor], device) -> torch.Tensor:
    assert len(tensors) <= _MAX_TENSORS, \
        f"too many tensors in one pp message ({len(tensors)})"
    meta = torch.zeros(_META_LEN, dtype=torch.int64)
    meta[0] = len(tensors)
    off = 1
    for t in tensors:
        meta[off] = t.dim()
        meta[off + 1] = _DTYPES.index(t.dtype)
        for i, sdim in enumerate(t.shape):
            meta[off + 2 + i] = sdim
        off += 2 + _MAX_DIMS
    return meta.to(device)


def _parse_meta(meta: torch.Tensor, device) -> List[torch.Tensor]:
    meta = meta.cpu().tolist()
    out = []
    off = 1
    for _ in range(meta[0]):
        ndim = meta[off]
        dtype = _DTYPES[meta[off + 1]]
        shape = meta[off + 2:off + 2 + ndim]
        off += 2 + _MAX_DIMS
        out.append(torch.empty(shape, dtype=dtype, device=device))
    return out


def _run_ops(ops):
    if ops:
        for w in dist.batch_isend_irecv(ops):
            w.wait()


def send_tensors(tensors: List[torch.Tensor], dst: int, device,
                 with_meta: bool = True):
    tensors = [t.contiguous().to(device) for t in tensors]
    ops = []
    if with_meta:
        ops.append(dist.P2POp(dist.isend, _meta_of(tensors, device), dst))
    for t in tensors:
        ops.append(dist.P2POp(dist.isend, t, dst))
    _run_ops(ops)


def recv_tensors(src: int, device) -> List[torch.Tensor]:
    meta = torch.zeros(_META_LEN, dtype=torch.int64, device=device)
    dist.recv(meta, src)
    bufs = _parse_meta(meta, device)
    ops = [dist.P2POp(dist.irecv, t, src) for t in bufs]
    _run_ops(ops)
    return bufs


def recv_into(bufs: List[torch.Tensor], src: int):
    """Receive payloads into preallocated buffers (no meta)."""
    ops = [dist.P2POp(dist.irecv, t, src) for t in bufs]
    _run_ops(ops)
    return bufs


def send_acts_recv_grads(acts: List[torch.Tensor],
                         grad_bufs: List[torch.Tensor], peer: int, device):
    """Fused steady-state op: send activations (with meta) to ``peer`` while
    receiving gradient payloads from ``peer``."""
    acts = [t.contiguous().to(device) for t in acts]
    ops = [dist.P2POp(dist.isend, _meta_of(acts, device), peer)]
    ops += [dist.P2POp(dist.isend, t, peer) for t in acts]
    ops += [dist.P2POp(dist.irecv, g, peer) for g in grad_bufs]
    _run_ops(ops)
    return grad_bufs
