"""Rotary position embedding (RoPE), fused q+k apply.

Replaces the reference's liger_rotary_pos_emb integration (ops/liger.py:66-70).
cos/sin tables are precomputed on host/device once (Appendix-B guidance:
on-device trig turns the op VALU-bound); the kernel is a pure
vectorized-elementwise pass over q and k.

Layout: q [b, s, h_q, d], k [b, s, h_k, d]; cos/sin [s, d/2] fp32.
Rotation is the "neox"/llama half-split: (x1, x2) -> (x1*c - x2*s, x2*c + x1*s)
with x1 = x[..., :d/2], x2 = x[..., d/2:].
"""
from typing import Tuple

import torch

from ._backend import dispatch


def build_rope_cache(seq_len: int, head_dim: int, base: float = 10000.0,
                     device=None, dtype=torch.float32
                     ) -> Tuple[torch.Tensor, torch.Tensor]:
    """``device=None`` follows the ambient default device (so model
    construction under ``with torch.device("cuda")`` keeps the tables on
    the same device as the parameters)."""
    inv_freq = 1.0 / (base ** (
        torch.arange(0, head_dim, 2, device=device, dtype=torch.float32)
        / head_dim))
    t = torch.arange(seq_len, device=device, dtype=torch.float32)
    freqs = torch.outer(t, inv_freq)  # [s, d/2]
    return freqs.cos().to(dtype), freqs.sin().to(dtype)


def _ref_apply(x, cos, sin):
    # x [b,s,h,d]; cos/sin [s,d/2]
    d2 = x.shape[-1] // 2
    x1 = x[..., :d2].float()
    x2 = x[..., d2:].float()
    c = cos[:x.shape[1]].view(1, -1, 1, d2)
    s = sin[:x.shape[1]].view(1, -1, 1, d2)
    o1 = x1 * c - x2 * s
    o2 = x2 * c + x1 * s
    return torch.cat([o1, o2], dim=-1).to(x.dtype)


class _RoPE(torch.autograd.Function):

    @staticmethod
    def forward(ctx, q, k, cos, sin):
        ext = dispatch(q)
        if q.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        q = q.contiguous()
        k = k.contiguous()
        # fp32 tables on q's device: a stale CPU (or dtype-cast) cache must
        # never reach the kernel as a host pointer / degraded precision
        cos = cos.to(q.device, torch.float32)
        sin = sin.to(q.device, torch.float32)
        if ext is not None:
            qo, ko = ext.rope_forward(q, k, cos, sin)
        else:
            qo, ko = _ref_apply(q, cos, sin), _ref_apply(k, cos, sin)
        ctx.save_for_backward(cos, sin)
        return qo, ko

    @staticmethod
    def backward(ctx, dq, dk):
        cos, sin = ctx.saved_tensors
        cos = cos.to(dq.device, torch.float32)
        sin = sin.to(dq.device, torch.float32)
        ext = dispatch(dq)
        if dq.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        dq = dq.contiguous()
        dk = dk.contiguous()
        if ext is not None:
            # inverse rotation = rotation by -theta  <=>  sin -> -sin
            dqo, dko = ext.rope_forward(dq, dk, cos, -sin)
        else:
            dqo = _ref_apply(dq, cos, -sin)
            dko = _ref_apply(dk, cos, -sin)
        return dqo, dko, None, None


def apply_rotary_pos_emb(q: torch.Tensor, k: torch.Tensor,
                         cos: torch.Tensor, sin: torch.Tensor):
    """q [b,s,hq,d], k [b,s,hk,d] -> rotated (q, k)."""
    return _RoPE.apply(q, k, cos, sin)
