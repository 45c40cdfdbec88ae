"""Fused cross-entropy (and fused-linear-cross-entropy).

Replaces the reference's LigerCrossEntropyLoss / lce_forward integration
(ops/liger.py:72-76). Two layers:

- :func:`cross_entropy`: HIP kernel computing per-row logsumexp + loss and,
  in backward, dlogits = (softmax - onehot) * scale without re-materializing
  softmax in a separate pass.
- :func:`linear_cross_entropy`: chunks the [N, hidden] input over N, runs
  matmul -> CE -> matmul-backward per chunk so the full [N, vocab] logits
  tensor is never materialized (what Liger's lce_forward buys, with the
  GEMMs on hipBLASLt and the CE math on the CDNA4 kernel — on GPU the
  logits chunk stays bf16 end to end).
"""
import torch

from ._backend import dispatch


class _CrossEntropy(torch.autograd.Function):

    @staticmethod
    def forward(ctx, logits, target, ignore_index):
        # logits [N, V], target [N]
        ext = dispatch(logits)
        if logits.dtype not in (torch.bfloat16, torch.float16):
            ext = None  # kernel is bf16-only; fp32 uses the composite path
        logits = logits.contiguous()
        if ext is not None:
            loss_sum, nvalid, lse = ext.cross_entropy_forward(
                logits, target, ignore_index)
        else:
            lf = logits.float()
            lse = torch.logsumexp(lf, dim=-1)
            valid = target != ignore_index
            nvalid = valid.sum()
            tgt = target.masked_fill(~valid, 0)
            picked = lf.gather(1, tgt.unsqueeze(1)).squeeze(1)
            loss_sum = ((lse - picked) * valid).sum()
        ctx.save_for_backward(logits, target, lse, nvalid)
        ctx.ignore_index = ignore_index
        n = nvalid.clamp_min(1)
        return loss_sum / n

    @staticmethod
    def backward(ctx, dloss):
        logits, target, lse, nvalid = ctx.saved_tensors
        ext = dispatch(logits)
        if logits.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        scale = dloss.float() / nvalid.clamp_min(1).float()
        if ext is not None:
            dlogits = ext.cross_entropy_backward(
                logits, target, lse, scale, ctx.ignore_index)
        else:
            lf = logits.float()
            soft = torch.exp(lf - lse.unsqueeze(1))
            valid = (target != ctx.ignore_index)
            tgt = target.masked_fill(~valid, 0)
            soft.scatter_add_(
                1, tgt.unsqueeze(1),
                -torch.ones_like(tgt, dtype=soft.dtype).unsqueeze(1))
            soft *= valid.unsqueeze(1)
            dlogits = (soft * scale).to(logits.dtype)
        return dlogits, None, None


def cross_entropy(logits: torch.Tensor, target: torch.Tensor,
                  ignore_index: int = -100) -> torch.Tensor:
    """Mean CE over rows whose target != ignore_index."""
    return _CrossEntropy.apply(logits, target, ignore_index)


def _chunk_ce_forward(ext, logits, target, ignore_index):
    """(loss_sum fp32, nvalid int, lse fp32[rows]) for one chunk."""
    if ext is not None:
        return ext.cross_entropy_forward(logits, target, ignore_index)
    lf = logits.float()
    lse = torch.logsumexp(lf, dim=-1)
    valid = target != ignore_index
    nvalid = valid.sum()
    tgt = target.masked_fill(~valid, 0)
    picked = lf.gather(1, tgt.unsqueeze(1)).squeeze(1)
    return ((lse - picked) * valid).sum(), nvalid, lse


def _chunk_ce_backward(ext, logits, target, lse, scale, ignore_index):
    """dlogits (logits dtype) for one chunk; scale is a 0-dim fp32 tensor."""
    if ext is not None:
        return ext.cross_entropy_backward(logits, target, lse, scale,
                                          ignore_index)
    lf = logits.float()
    soft = torch.exp(lf - lse.unsqueeze(1))
    valid = target != ignore_index
    tgt = target.masked_fill(~valid, 0)
    soft.scatter_add_(
        1, tgt.unsqueeze(1),
        -torch.ones_like(tgt, dtype=soft.dtype).unsqueeze(1))
    soft *= valid.unsqueeze(1)
    return (soft * scale).to(logits.dtype)


def _ce_gemm(ext, a, b, ta, tb):
    """Chunk GEMM on the tuned hipBLASLt path (bf16 GPU), else torch."""
    if ext is not None and a.is_cuda and a.dtype == torch.bfloat16:
        from .linear import _algo_for
        m = a.shape[1] if ta else a.shape[0]
        k = a.shape[0] if ta else a.shape[1]
        n = b.shape[0] if tb else b.shape[1]
        a = a if a.is_contiguous() else a.contiguous()
        return ext.lt_gemm(a, b, ta, tb, _algo_for(m, n, k, ta, tb))
    op_a = a.t() if ta else a
    op_b = b.t() if tb else b
    return (op_a @ op_b).contiguous()


class _LinearCrossEntropy(torch.autograd.Function):
    """y = CE(x @ W^T, target) without materializing full logits.

    x [N, H], W [V, H]. Forward runs in row chunks (GEMM -> CE kernel, lse
    kept per row); backward recomputes each chunk's logits (cheap GEMM) and
    accumulates dx and dW.
    """

    CHUNK = 8192

    @staticmethod
    def forward(ctx, x, weight, target, ignore_index):
        N = x.shape[0]
        ext = dispatch(x)
        if x.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        total = x.new_zeros((), dtype=torch.float32)
        nvalid_t = torch.zeros((), dtype=torch.long, device=x.device)
        lse_all = torch.empty(N, dtype=torch.float32, device=x.device)
        for s in range(0, N, _LinearCrossEntropy.CHUNK):
            e = min(N, s + _LinearCrossEntropy.CHUNK)
            logits = _ce_gemm(ext, x[s:e], weight, False, True)
            loss_sum, nvalid, lse = _chunk_ce_forward(
                ext, logits, target[s:e], ignore_index)
            total += loss_sum.float()
            nvalid_t += nvalid.long()
            lse_all[s:e] = lse
        ctx.save_for_backward(x, weight, target, lse_all, nvalid_t)
        ctx.ignore_index = ignore_index
        return total / nvalid_t.clamp_min(1).float()

    @staticmethod
    def backward(ctx, dloss):
        x, weight, target, lse_all, nvalid_t = ctx.saved_tensors
        ignore_index = ctx.ignore_index
        ext = dispatch(x)
        if x.dtype not in (torch.bfloat16, torch.float16):
            ext = None
        N = x.shape[0]
        scale = (dloss.float() / nvalid_t.clamp_min(1).float())
        dx = torch.empty_like(x)
        dw = torch.zeros_like(weight, dtype=torch.float32)
        for s in range(0, N, _LinearCrossEntropy.CHUNK):
            e = min(N, s + _LinearCrossEntropy.CHUNK)
            logits = _ce_gemm(ext, x[s:e], weight, False, True)
            dl = _chunk_ce_backward(ext, logits, target[s:e], lse_all[s:e],
                                    scale, ignore_index)
            dx[s:e] = _ce_gemm(ext, dl, weight, False, False)
            dw += _ce_gemm(ext, dl, x[s:e], True, False).float()
        return dx, dw.to(weight.dtype), None, None


def linear_cross_entropy(x: torch.Tensor, weight: torch.Tensor,
                         target: torch.Tensor,
                         ignore_index: int = -100) -> torch.Tensor:
    return _LinearCrossEntropy.apply(x, weight, target, ignore_index)
