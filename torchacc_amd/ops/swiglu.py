"""SwiGLU activation: silu(gate) * up, fused fwd/bwd.

Replaces the reference's LigerSiLUMulFunction (ops/liger.py:21-28). Pure
bandwidth-bound elementwise op; the HIP kernel is short8-vectorized bf16.
"""
import torch

from ._backend import dispatch


class _SwiGLU(torch.autograd.Function):

    @staticmethod
    def forward(ctx, gate, up):
        ext = dispatch(gate)
        if gate.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        gate = gate.contiguous()
        up = up.contiguous()
        ctx.save_for_backward(gate, up)
        if ext is not None:
            return ext.swiglu_forward(gate, up)
        g = gate.float()
        return (g * torch.sigmoid(g) * up.float()).to(gate.dtype)

    @staticmethod
    def backward(ctx, dy):
        gate, up = ctx.saved_tensors
        ext = dispatch(gate)
        if gate.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        dy = dy.contiguous()
        if ext is not None:
            dgate, dup = ext.swiglu_backward(dy, gate, up)
            return dgate, dup
        g = gate.float()
        u = up.float()
        d = dy.float()
        sig = torch.sigmoid(g)
        silu = g * sig
        dgate = (d * u * (sig * (1 + g * (1 - sig)))).to(gate.dtype)
        dup = (d * silu).to(up.dtype)
        return dgate, dup


def swiglu(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    return _SwiGLU.apply(gate, up)
