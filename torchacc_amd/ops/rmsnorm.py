"""RMSNorm with a hand-written CDNA4 kernel (csrc/rmsnorm.hip).

Replaces the reference's Liger Triton RMSNorm (ops/liger.py:10-18). Forward
saves the inverse RMS (fp32) for backward. CPU path is the fp32 reference the
numerics tests compare against.
"""
import torch

from ._backend import dispatch


def _ref_rms_forward(x, weight, eps):
    xf = x.float()
    inv_rms = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    y = (xf * inv_rms) * weight.float()
    return y.to(x.dtype), inv_rms


class _RMSNorm(torch.autograd.Function):

    @staticmethod
    def forward(ctx, x, weight, eps):
        ext = dispatch(x)
        if x.dtype not in (torch.bfloat16, torch.float16, torch.float32):
            ext = None
        x = x.contiguous()
        if ext is not None:
            y, inv_rms = ext.rmsnorm_forward(x, weight, eps)
        else:
            y, inv_rms = _ref_rms_forward(x, weight, eps)
            inv_rms = inv_rms.squeeze(-1)
        ctx.save_for_backward(x, weight, inv_rms)
        ctx.eps = eps
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, inv_rms = ctx.saved_tensors
        ext = dispatch(x)
        if x.dtype not in (torch.bfloat16, torch.float16):
            ext = None  # kernel backward is bf16-only
        dy = dy.contiguous()
        if ext is not None:
            dx, dw = ext.rmsnorm_backward(dy, x, weight, inv_rms)
        else:
            xf = x.float()
            dyf = dy.float()
            wf = weight.float()
            r = inv_rms.unsqueeze(-1)
            xhat = xf * r
            wdy = dyf * wf
            ddot = (wdy * xhat).mean(-1, keepdim=True)
            dx = (r * (wdy - xhat * ddot)).to(x.dtype)
            dw = (dyf * xhat).reshape(-1, x.shape[-1]).sum(0).to(weight.dtype)
        return dx, dw, None


def rms_norm(x: torch.Tensor, weight: torch.Tensor,
             eps: float = 1e-6) -> torch.Tensor:
    return _RMSNorm.apply(x, weight, eps)


class RMSNorm(torch.nn.Module):

    def __init__(self, hidden_size: int, eps: float = 1e-6):
        super().__init__()
        self.weight = torch.nn.Parameter(torch.ones(hidden_size))
        self.variance_epsilon = eps

    def forward(self, x):
        return rms_norm(x, self.weight, self.variance_epsilon)

    def extra_repr(self):
        return f"{self.weight.shape[0]}, eps={self.variance_epsilon}"


class _AddRMSNorm(torch.autograd.Function):
    """Fused residual add + RMSNorm: r = x + residual; y = rmsnorm(r) * w.
    Returns (y, r); the residual stream r flows to the next layer so its
    incoming grad is fused into dx in backward (one kernel instead of an
    add + a norm + an elementwise grad-add)."""

    @staticmethod
    def forward(ctx, x, residual, weight, eps):
        ext = dispatch(x)
        if x.dtype not in (torch.bfloat16, torch.float16):
            ext = None  # composite path for fp32-on-GPU
        x = x.contiguous()
        has_resid = residual is not None
        if ext is not None:
            y, r, inv_rms = ext.add_rmsnorm_forward(
                x, residual.contiguous() if has_resid else
                torch.empty(0, device=x.device, dtype=x.dtype),
                weight, eps)
        else:
            r = (x + residual) if has_resid else x
            y, inv_rms = _ref_rms_forward(r, weight, eps)
            inv_rms = inv_rms.squeeze(-1)
        ctx.save_for_backward(r, weight, inv_rms)
        ctx.eps = eps
        ctx.has_resid = has_resid
        return y, r

    @staticmethod
    def backward(ctx, dy, dresid):
        r, weight, inv_rms = ctx.saved_tensors
        ext = dispatch(r)
        if r.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        dy = dy.contiguous()
        if ext is not None:
            dx, dw = ext.add_rmsnorm_backward(
                dy, dresid.contiguous() if dresid is not None else
                torch.empty(0, device=r.device, dtype=r.dtype),
                r, weight, inv_rms)
        else:
            rf = r.float()
            dyf = dy.float()
            wf = weight.float()
            ri = inv_rms.unsqueeze(-1)
            xhat = rf * ri
            wdy = dyf * wf
            ddot = (wdy * xhat).mean(-1, keepdim=True)
            dx = (ri * (wdy - xhat * ddot))
            if dresid is not None:
                dx = dx + dresid.float()
            dx = dx.to(r.dtype)
            dw = (dyf * xhat).reshape(-1, r.shape[-1]).sum(0).to(
                weight.dtype)
        dres = dx if ctx.has_resid else None
        return dx, dres, dw, None


def fused_add_rms_norm(x: torch.Tensor, residual, weight: torch.Tensor,
                       eps: float = 1e-6):
    """(y, new_residual) = (rmsnorm(x + residual) * w, x + residual);
    residual may be None (plain norm, r = x)."""
    return _AddRMSNorm.apply(x, residual, weight, eps)
