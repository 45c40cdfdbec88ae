"""Ring attention (blockwise attention with KV rotating around the ring).

Reimplements the semantics of the reference's ring_attn.py:22-508 (itself
zhuzilin/ring-flash-attention-derived) for the eager MI355X backend:

- forward: ``cp`` steps; each step overlaps the async neighbor KV exchange
  (RCCL isend/irecv over xGMI) with the local flash-attention call on the
  block currently held, merging (out, lse) with the stable online-softmax
  update. The sequence is distributed in rank order (rank r holds tokens
  [r*s_local, (r+1)*s_local)); with ``causal=True`` the block from source
  rank src needs: local causal attention when src == rank, full attention
  when src < rank, and no compute when src > rank.
- backward: KV ring a second time in the same direction while a (dk, dv)
  accumulator travels with each block; each step computes the per-block
  gradients against the GLOBAL (out, lse) — p is normalized by the final
  lse and delta = rowsum(dout*out) — so contributions sum exactly; dq
  accumulates locally.

The reference's own ring/2D tests were skipped for correctness issues
(test_context_parallel.py:104-109); this implementation is validated against
single-device flash attention in tests/ops/test_context_parallel.py.
"""
import torch
import torch.distributed as dist

from .._backend import dispatch
from ..flash_attn import _ref_attention, _ref_fa_backward
from .init_group import get_inter_cp_group
from .utils import RingComm, update_out_and_lse


def _block_fwd(q, k, v, softmax_scale, causal):
    ext = dispatch(q)
    if ext is not None:
        return ext.fa_forward(q, k, v, softmax_scale, causal, -1, -1,
                              torch.empty(0), torch.empty(0),
                              torch.empty(0), 0.0, 0)
    return _ref_attention(q, k, v, softmax_scale, causal, (-1, -1))


def _block_bwd(dout, q, k, v, out, lse, softmax_scale, causal):
    ext = dispatch(q)
    if ext is not None:
        return ext.fa_backward(dout, q, k, v, out, lse, softmax_scale,
                               causal, -1, -1, torch.empty(0),
                               torch.empty(0), torch.empty(0), 0.0, 0)
    return _ref_fa_backward(dout, q, k, v, out, lse, softmax_scale, causal,
                            (-1, -1), None, None)


def ring_flash_attn_forward(group, q, k, v, softmax_scale, causal=True):
    comm = RingComm(group)
    rank, ws = comm.rank, comm.ws
    out = None
    lse = None
    kv = torch.stack([k, v])  # single message per step: [2,b,s,hk,d]
    next_kv = None
    for step in range(ws):
        if step + 1 < ws:
            next_kv = comm.send_recv(kv)
            comm.commit()
        src = (rank - step) % ws
        if (not causal) or src <= rank:
            blk_out, blk_lse = _block_fwd(q, kv[0], kv[1], softmax_scale,
                                          causal and src == rank)
            out, lse = update_out_and_lse(out, lse, blk_out, blk_lse)
        if step + 1 < ws:
            comm.wait()
            kv = next_kv
    assert out is not None
    return out.to(q.dtype), lse


def ring_flash_attn_backward(group, dout, q, k, v, out, lse, softmax_scale,
                             causal=True):
    kv_comm = RingComm(group)
    d_comm = RingComm(group)
    rank, ws = kv_comm.rank, kv_comm.ws
    dq = torch.zeros_like(q, dtype=torch.float32)
    kv = torch.stack([k, v])
    dkv_acc = torch.zeros(kv.shape, dtype=torch.float32, device=q.device)
    next_kv = None
    next_dkv = None
    for step in range(ws):
        if step + 1 < ws:
            next_kv = kv_comm.send_recv(kv)
            kv_comm.commit()
        src = (rank - step) % ws
        if (not causal) or src <= rank:
            bdq, bdk, bdv = _block_bwd(dout, q, kv[0].contiguous(),
                                       kv[1].contiguous(), out, lse,
                                       softmax_scale,
                                       causal and src == rank)
            dq += bdq.float()
            dkv_acc[0] += bdk.float()
            dkv_acc[1] += bdv.float()
        # accumulator follows its block to the next rank
        next_dkv = d_comm.send_recv(dkv_acc)
        d_comm.commit()
        d_comm.wait()
        dkv_acc = next_dkv
        if step + 1 < ws:
            kv_comm.wait()
            kv = next_kv
    # after ws hops the accumulator arriving holds grads for MY block
    return (dq.to(q.dtype), dkv_acc[0].to(k.dtype), dkv_acc[1].to(v.dtype))


class RingFlashAttnFunc(torch.autograd.Function):

    @staticmethod
    def forward(ctx, q, k, v, softmax_scale, causal, group):
        if softmax_scale is None:
            softmax_scale = q.shape[-1] ** (-0.5)
        q, k, v = [t.contiguous() for t in (q, k, v)]
        out, lse = ring_flash_attn_forward(group, q, k, v, softmax_scale,
                                           causal)
        ctx.save_for_backward(q, k, v, out, lse)
        ctx.softmax_scale = softmax_scale
        ctx.causal = causal
        ctx.group = group
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse = ctx.saved_tensors
        dq, dk, dv = ring_flash_attn_backward(
            ctx.group, dout.contiguous(), q, k, v, out, lse,
            ctx.softmax_scale, ctx.causal)
        return dq, dk, dv, None, None, None


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   softmax_scale=None, causal: bool = True,
                   process_group=None) -> torch.Tensor:
    """User API (reference ring_attn.py:431-508): q/k/v [b, s_local, h, d]
    sequence-sharded over the ring group in rank order."""
    group = process_group if process_group is not None \
        else get_inter_cp_group()
    if group is None or dist.get_world_size(group) == 1:
        from ..flash_attn import flash_attn_xla
        return flash_attn_xla(q, k, v, softmax_scale=softmax_scale,
                              causal=causal)
    return RingFlashAttnFunc.apply(q, k, v, softmax_scale, causal, group)


def ring_flash_attn_qkvpacked_func(qkv, softmax_scale=None, causal=True,
                                   process_group=None):
    q, k, v = qkv.unbind(2)
    return ring_attention(q, k, v, softmax_scale=softmax_scale,
                          causal=causal, process_group=process_group)


def ring_flash_attn_kvpacked_func(q, kv, softmax_scale=None, causal=True,
                                  process_group=None):
    k, v = kv.unbind(2)
    return ring_attention(q, k, v, softmax_scale=softmax_scale,
                          causal=causal, process_group=process_group)
