"""Context-parallel building blocks (reference ops/context_parallel/utils.py).

- differentiable sequence split/gather (:175-259)
- differentiable all-to-all scattering one dim / gathering another (:262-299)
- numerically-stable online-softmax (out, lse) merge (:302-343)
- RingComm: neighbor KV exchange via batched isend/irecv with even/odd op
  ordering to avoid deadlock (:368-423), RCCL-native on MI355X.
"""
from typing import List, Optional, Tuple

import torch
import torch.distributed as dist


# ---------------------------------------------------------------------------
# split / gather
# ---------------------------------------------------------------------------

class SplitForwardGatherBackward(torch.autograd.Function):
    """Forward: keep this rank's chunk along ``dim``; backward: all-gather
    the grads along ``dim``."""

    @staticmethod
    def forward(ctx, x, dim, group, grad_scale):
        ctx.dim = dim
        ctx.group = group
        ctx.grad_scale = grad_scale
        ws = dist.get_world_size(group)
        rank = dist.get_rank(group)
        ctx.ws = ws
        chunks = x.chunk(ws, dim=dim)
        return chunks[rank].contiguous()

    @staticmethod
    def backward(ctx, grad):
        ws = ctx.ws
        grad = grad.contiguous()
        parts = [torch.empty_like(grad) for _ in range(ws)]
        dist.all_gather(parts, grad, group=ctx.group)
        out = torch.cat(parts, dim=ctx.dim)
        if ctx.grad_scale == "up":
            out = out * ws
        elif ctx.grad_scale == "down":
            out = out / ws
        return out, None, None, None


class GatherForwardSplitBackward(torch.autograd.Function):
    """Forward: all-gather along ``dim``; backward: keep this rank's chunk."""

    @staticmethod
    def forward(ctx, x, dim, group, grad_scale):
        ctx.dim = dim
        ctx.group = group
        ctx.grad_scale = grad_scale
        ws = dist.get_world_size(group)
        ctx.rank = dist.get_rank(group)
        ctx.ws = ws
        x = x.contiguous()
        parts = [torch.empty_like(x) for _ in range(ws)]
        dist.all_gather(parts, x, group=group)
        return torch.cat(parts, dim=dim)

    @staticmethod
    def backward(ctx, grad):
        chunk = grad.chunk(ctx.ws, dim=ctx.dim)[ctx.rank].contiguous()
        if ctx.grad_scale == "up":
            chunk = chunk * ctx.ws
        elif ctx.grad_scale == "down":
            chunk = chunk / ctx.ws
        return chunk, None, None, None


def split_forward_gather_backward(x, dim, group, grad_scale=None):
    return SplitForwardGatherBackward.apply(x, dim, group, grad_scale)


def gather_forward_split_backward(x, dim, group, grad_scale=None):
    return GatherForwardSplitBackward.apply(x, dim, group, grad_scale)


# ---------------------------------------------------------------------------
# all-to-all
# ---------------------------------------------------------------------------

class AllToAll(torch.autograd.Function):
    """Differentiable all-to-all: scatter ``scatter_dim``, gather
    ``gather_dim`` (reference utils.py:262-299)."""

    @staticmethod
    def forward(ctx, x, scatter_dim, gather_dim, group):
        ctx.scatter_dim = scatter_dim
        ctx.gather_dim = gather_dim
        ctx.group = group
        return all_to_all(x, scatter_dim, gather_dim, group)

    @staticmethod
    def backward(ctx, grad):
        return (all_to_all(grad.contiguous(), ctx.gather_dim,
                           ctx.scatter_dim, ctx.group), None, None, None)


def all_to_all(x: torch.Tensor, scatter_dim: int, gather_dim: int,
               group) -> torch.Tensor:
    """Non-differentiable a2a: split x into ws chunks along scatter_dim,
    exchange, concatenate received chunks along gather_dim."""
    ws = dist.get_world_size(group)
    if ws == 1:
        return x
    inputs = [c.contiguous() for c in x.chunk(ws, dim=scatter_dim)]
    outputs = [torch.empty_like(c) for c in inputs]
    if dist.get_backend(group) == "gloo":
        # gloo has no alltoall: emulate with batched isend/irecv
        rank = dist.get_rank(group)
        ranks = dist.get_process_group_ranks(group)
        ops = []
        for j in range(ws):
            if j == rank:
                outputs[j].copy_(inputs[j])
                continue
            ops.append(dist.P2POp(dist.isend, inputs[j], ranks[j],
                                  group=group))
            ops.append(dist.P2POp(dist.irecv, outputs[j], ranks[j],
                                  group=group))
        if ops:
            for w in dist.batch_isend_irecv(ops):
                w.wait()
    else:
        dist.all_to_all(outputs, inputs, group=group)
    return torch.cat(outputs, dim=gather_dim)


def diff_all_to_all(x, scatter_dim, gather_dim, group):
    return AllToAll.apply(x, scatter_dim, gather_dim, group)


# ---------------------------------------------------------------------------
# online-softmax merge
# ---------------------------------------------------------------------------

def update_out_and_lse(out: Optional[torch.Tensor],
                       lse: Optional[torch.Tensor],
                       block_out: torch.Tensor,
                       block_lse: torch.Tensor
                       ) -> Tuple[torch.Tensor, torch.Tensor]:
    """Merge a new attention block into the running (out, lse).

    out/block_out: [b, s, h, d] (any dtype); lse/block_lse: [b, h, s] fp32.
    logaddexp form (equivalent to the reference's sigmoid/logsigmoid update,
    utils.py:302-343, but safe when either side is -inf — a fully-masked
    varlen block contributes lse = -inf and must be a no-op):
        new_lse = logaddexp(lse, block_lse)
        out     = out + exp(block_lse - new_lse) * (block_out - out)
    """
    if out is None:
        return block_out.float(), block_lse
    blk = block_out.float()
    new_lse = torch.logaddexp(lse, block_lse)
    w = torch.exp(block_lse - new_lse)             # [b,h,s], 0 for -inf blk
    w = torch.nan_to_num(w)                        # both -inf: dead rows
    out = out + w.transpose(-2, -1).unsqueeze(-1) * (blk - out)
    return out, new_lse


# ---------------------------------------------------------------------------
# ring communication
# ---------------------------------------------------------------------------

class RingComm:
    """Neighbor exchange on a ring over the CP group: each call sends a
    tensor to rank+1 and receives from rank-1 (batched isend/irecv; even
    ranks post send first, odd ranks recv first — deadlock-free on both
    gloo and RCCL)."""

    def __init__(self, group):
        self.group = group
        self.rank = dist.get_rank(group)
        self.ws = dist.get_world_size(group)
        ranks = dist.get_process_group_ranks(group) if group is not None \
            else list(range(dist.get_world_size()))
        self.send_rank = ranks[(self.rank + 1) % self.ws]
        self.recv_rank = ranks[(self.rank - 1) % self.ws]
        self._ops: List[dist.P2POp] = []
        self._reqs = None

    def send_recv(self, to_send: torch.Tensor,
                  recv_buf: Optional[torch.Tensor] = None) -> torch.Tensor:
        if recv_buf is None:
            recv_buf = torch.empty_like(to_send)
        send_op = dist.P2POp(dist.isend, to_send.contiguous(),
                             self.send_rank, group=self.group)
        recv_op = dist.P2POp(dist.irecv, recv_buf, self.recv_rank,
                             group=self.group)
        if self.rank % 2 == 0:
            self._ops += [send_op, recv_op]
        else:
            self._ops += [recv_op, send_op]
        return recv_buf

    def commit(self):
        assert self._reqs is None, "commit called twice without wait"
        if self._ops:
            self._reqs = dist.batch_isend_irecv(self._ops)
        else:
            self._reqs = []
        self._ops = []

    def wait(self):
        assert self._reqs is not None, "wait called before commit"
        for r in self._reqs:
            r.wait()
        self._reqs = None
