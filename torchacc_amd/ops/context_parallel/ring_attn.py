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
- varlen: global per-batch true lengths (reference ring_attention k_lens
  math, ring_attn.py:431-508) are clamped into each source block —
  blk_k_lens = clamp(k_lens - src*s_local, 0, s_local) — and handed to the
  varlen-by-lens kernels, so padded tails contribute nothing anywhere on
  the ring. Blocks that are entirely padding are skipped.
- backward: KV ring a second time in the same direction while a (dk, dv)
  fp32 accumulator travels with each block; each step computes the
  per-block gradients against the GLOBAL (out, lse) — p is normalized by
  the final lse and delta = rowsum(dout*out) — so contributions sum
  exactly; dq accumulates locally. The accumulator exchange is committed
  right after each step's fold and awaited only at the NEXT step's fold,
  overlapping the wire time with the next block's backward kernels (the
  reference serialized this exchange inside every step).

The reference's own ring/2D tests were skipped for correctness issues
(test_context_parallel.py:104-109); this implementation is validated against
single-device flash attention in tests/ops/test_context_parallel.py.
"""
from typing import Optional

import torch
import torch.distributed as dist

from .._backend import dispatch
from ..flash_attn import _ref_attention, _ref_fa_backward
from .init_group import get_inter_cp_group
from .utils import RingComm, update_out_and_lse

_E = torch.empty(0)


def _block_fwd(q, k, v, softmax_scale, causal, q_lens=None, k_lens=None):
    ext = dispatch(q)
    if ext is not None:
        return ext.fa_forward(
            q, k, v, softmax_scale, causal, -1, -1,
            q_lens if q_lens is not None else _E,
            k_lens if k_lens is not None else _E, _E, 0.0, 0)
    return _ref_attention(q, k, v, softmax_scale, causal, (-1, -1),
                          q_lens, k_lens)


def _block_bwd(dout, q, k, v, out, lse, softmax_scale, causal, q_lens=None,
               k_lens=None):
    ext = dispatch(q)
    if ext is not None:
        return ext.fa_backward(
            dout, q, k, v, out, lse, softmax_scale, causal, -1, -1,
            q_lens if q_lens is not None else _E,
            k_lens if k_lens is not None else _E, _E, 0.0, 0)
    return _ref_fa_backward(dout, q, k, v, out, lse, softmax_scale, causal,
                            (-1, -1), q_lens, k_lens)


def _block_lens(k_lens_cpu, src, s_local, device):
    """Per-batch valid-key counts inside source rank ``src``'s block, or
    None when the whole block is valid; False when entirely padding."""
    if k_lens_cpu is None:
        return None, True
    blk = (k_lens_cpu - src * s_local).clamp(0, s_local)
    if int(blk.max()) == 0:
        return None, False
    if int(blk.min()) == s_local:
        return None, True
    return blk.to(device=device, dtype=torch.int32), True

def ring_flash_attn_forward(group, q, k, v, softmax_scale, causal=True,
                            q_lens=None, k_lens=None):
    comm = RingComm(group)
    rank, ws = comm.rank, comm.ws
    s_local = k.shape[1]
    k_lens_cpu = k_lens.cpu() if k_lens is not None else None
    my_q_lens, _ = _block_lens(
        q_lens.cpu() if q_lens is not None else None, rank, q.shape[1],
        q.device)
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
            blk_k_lens, any_valid = _block_lens(k_lens_cpu, src, s_local,
                                                q.device)
            if any_valid:
                blk_out, blk_lse = _block_fwd(
                    q, kv[0], kv[1], softmax_scale,
                    causal and src == rank, my_q_lens, blk_k_lens)
                if blk_k_lens is not None:
                    # batches with zero valid keys in THIS block must not
                    # contribute to the merge regardless of how the kernel
                    # fills fully-masked rows
                    zb = blk_k_lens == 0
                    if bool(zb.any()):
                        blk_lse = blk_lse.masked_fill(
                            zb.view(-1, 1, 1), float("-inf"))
                        blk_out = blk_out * (~zb).view(-1, 1, 1, 1).to(
                            blk_out.dtype)
                out, lse = update_out_and_lse(out, lse, blk_out, blk_lse)
        if step + 1 < ws:
            comm.wait()
            kv = next_kv
    assert out is not None
    if my_q_lens is not None or k_lens_cpu is not None:
        # rows beyond the true query length (and any residual -inf/NaN from
        # rows with zero valid keys): defined as zero out / zero lse,
        # matching _ref_attention's qmask semantics. masked_fill, not
        # multiply — NaN times zero is still NaN.
        if my_q_lens is not None:
            pos = torch.arange(q.shape[1], device=q.device)
            dead = pos.unsqueeze(0) >= my_q_lens.unsqueeze(1)  # [b,s]
        else:
            dead = torch.zeros(q.shape[0], q.shape[1], dtype=torch.bool,
                               device=q.device)
        dead = dead | torch.isnan(lse[:, 0, :]) | \
            torch.isinf(lse[:, 0, :])
        out = out.masked_fill(dead.unsqueeze(-1).unsqueeze(-1), 0)
        lse = lse.masked_fill(dead.unsqueeze(1), 0)            # [b,h,s]
    return out.to(q.dtype), lse


def ring_flash_attn_backward(group, dout, q, k, v, out, lse, softmax_scale,
                             causal=True, q_lens=None, k_lens=None):
    kv_comm = RingComm(group)
    d_comm = RingComm(group)
    rank, ws = kv_comm.rank, kv_comm.ws
    s_local = k.shape[1]
    k_lens_cpu = k_lens.cpu() if k_lens is not None else None
    my_q_lens, _ = _block_lens(
        q_lens.cpu() if q_lens is not None else None, rank, q.shape[1],
        q.device)
    if my_q_lens is not None:
        # contributions from padded query rows must vanish everywhere
        pos = torch.arange(q.shape[1], device=q.device)
        qmask = pos.unsqueeze(0) < my_q_lens.unsqueeze(1)
        dout = dout * qmask.unsqueeze(-1).unsqueeze(-1).to(dout.dtype)
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
        bdk = bdv = None
        if (not causal) or src <= rank:
            blk_k_lens, any_valid = _block_lens(k_lens_cpu, src, s_local,
                                                q.device)
            if any_valid:
                bdq, bdk, bdv = _block_bwd(
                    dout, q, kv[0].contiguous(), kv[1].contiguous(), out,
                    lse, softmax_scale, causal and src == rank, my_q_lens,
                    blk_k_lens)
                if blk_k_lens is not None:
                    zb = blk_k_lens == 0
                    if bool(zb.any()):
                        z4 = zb.view(-1, 1, 1, 1)
                        bdq = bdq.masked_fill(z4, 0)
                        bdk = bdk.masked_fill(z4, 0)
                        bdv = bdv.masked_fill(z4, 0)
                dq += bdq.float()
        # the accumulator traveling with this block arrives while the
        # block's backward kernels run (committed last step, awaited here)
        if step > 0:
            d_comm.wait()
            dkv_acc = next_dkv
        if bdk is not None:
            dkv_acc[0] += bdk.float()
            dkv_acc[1] += bdv.float()
        # send it onward with the block; overlap with next step's compute
        next_dkv = d_comm.send_recv(dkv_acc)
        d_comm.commit()
        if step + 1 < ws:
            kv_comm.wait()
            kv = next_kv
    # after ws hops the accumulator arriving holds grads for MY block
    d_comm.wait()
    dkv_acc = next_dkv
    return (dq.to(q.dtype), dkv_acc[0].to(k.dtype), dkv_acc[1].to(v.dtype))


class RingFlashAttnFunc(torch.autograd.Function):

    @staticmethod
    def forward(ctx, q, k, v, softmax_scale, causal, group, q_lens, k_lens):
        if softmax_scale is None:
            softmax_scale = q.shape[-1] ** (-0.5)
        q, k, v = [t.contiguous() for t in (q, k, v)]
        out, lse = ring_flash_attn_forward(group, q, k, v, softmax_scale,
                                           causal, q_lens, k_lens)
        ctx.save_for_backward(
            q, k, v, out, lse,
            q_lens if q_lens is not None else _E,
            k_lens if k_lens is not None else _E)
        ctx.softmax_scale = softmax_scale
        ctx.causal = causal
        ctx.group = group
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse, q_lens, k_lens = ctx.saved_tensors
        dq, dk, dv = ring_flash_attn_backward(
            ctx.group, dout.contiguous(), q, k, v, out, lse,
            ctx.softmax_scale, ctx.causal,
            q_lens if q_lens.numel() else None,
            k_lens if k_lens.numel() else None)
        return dq, dk, dv, None, None, None, None, None


def ring_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                   softmax_scale=None, causal: bool = True,
                   process_group=None,
                   q_lens: Optional[torch.Tensor] = None,
                   k_lens: Optional[torch.Tensor] = None) -> torch.Tensor:
    """User API (reference ring_attn.py:431-508): q/k/v [b, s_local, h, d]
    sequence-sharded over the ring group in rank order. ``q_lens``/
    ``k_lens`` are GLOBAL per-batch true lengths [b]; the per-rank block
    lengths are derived internally (the reference's true_k_lens math)."""
    group = process_group if process_group is not None \
        else get_inter_cp_group()
    if group is None or dist.get_world_size(group) == 1:
        from ..flash_attn import FlashAttnFunc
        if softmax_scale is None:
            softmax_scale = q.shape[-1] ** (-0.5)
        out, _ = FlashAttnFunc.apply(
            q, k, v, 0.0, softmax_scale, causal, (-1, -1), None, False,
            q_lens.to(torch.int32) if q_lens is not None else None,
            k_lens.to(torch.int32) if k_lens is not None else None)
        return out
    if q_lens is not None:
        q_lens = q_lens.to(torch.int32)
    if k_lens is not None:
        k_lens = k_lens.to(torch.int32)
    return RingFlashAttnFunc.apply(q, k, v, softmax_scale, causal, group,
                                   q_lens, k_lens)


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
