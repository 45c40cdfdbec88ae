"""Flash-attention v2 ops on hand-written CDNA4 MFMA kernels.

Reimplements the API surface of the reference's ops/flash_attn.py:11-601
(FlashAttnXla / FlashAttnVarlenXla / FlashAttnVarlenPositionIdsXla and the
user wrappers flash_attn_xla / flash_attn_varlen_xla /
flash_attn_varlen_position_ids_xla) on a single eager ROCm backend:

- fixed-length [b, s, h, d] fwd/bwd with causal, GQA (h_k | h) and sliding
  window;
- varlen via an int32 attention mask [b, s_k] (right-padding; per-batch
  lengths are extracted and passed to the kernel);
- packed-sequence varlen via position_ids [1, total] (cu_seqlens derived
  from position-id resets, reference ops/flash_attn.py:173-218) — used by
  the HF packed-sample training path;
- explicit cu_seqlens varlen (used by ring attention).

The softmax log-sum-exp is returned in fp32 [b, h, s] exactly like FA2 so the
context-parallel LSE merge math is unchanged. bf16 on the CDNA4 kernels
(other dtypes/head dims run a composite fallback); alibi and dropout
dispatch to the full-featured kernel family (csrc/flash_attn_extra_*.hip).
"""
import math
import threading
from collections import deque

import torch

from ._backend import dispatch

# ---- selective activation checkpointing (Megatron-style) -------------------
# utils/checkpoint.py passes sac_contexts as torch.utils.checkpoint's
# context_fn: the capture context retains (out, lse, seed) of every
# attention call inside the checkpointed region, and the recompute context
# replays them so recomputation skips the attention kernels entirely.

_sac_tls = threading.local()


class _SacContext:

    def __init__(self, mode, queue):
        self.mode = mode
        self.queue = queue

    def __enter__(self):
        self.prev = getattr(_sac_tls, "state", None)
        _sac_tls.state = (self.mode, self.queue)

    def __exit__(self, *exc):
        _sac_tls.state = self.prev
        return False


def sac_contexts():
    """Fresh queue per checkpoint invocation: capture/recompute pairs stay
    matched even under PP micro-batch interleaving."""
    q = deque()
    return _SacContext("capture", q), _SacContext("recompute", q)


def _sac_pop():
    state = getattr(_sac_tls, "state", None)
    if state is not None and state[0] == "recompute" and state[1]:
        return state[1].popleft()
    return None


def _sac_push(entry):
    state = getattr(_sac_tls, "state", None)
    if state is not None and state[0] == "capture":
        state[1].append(entry)


_warned_dims = set()


def _pad_target(d: int):
    """Kernel head_dim to zero-pad to (reference pads odd head dims and
    trims in backward, ops/flash_attn.py:166-168); None -> composite."""
    if d in (64, 128):
        return None  # native
    if d < 64:
        return 64
    if d < 128:
        return 128
    return -1  # unsupported: composite


def _kernel_ext(q):
    """The CDNA4 kernels cover bf16 AND fp16 (AttnElem-templated MFMA
    ladder) with head_dim 64/128 natively and any d <= 128 via zero
    padding; d > 128 runs the fp32 composite (warned once per dim — the
    flagship models are all head_dim 128)."""
    ext = dispatch(q)
    if ext is None:
        return None
    if q.dtype not in (torch.bfloat16, torch.float16):
        return None
    d = q.shape[-1]
    if _pad_target(d) == -1:
        if d not in _warned_dims:
            _warned_dims.add(d)
            from ..utils.logger import logger
            logger.warning(
                "flash-attention head_dim %d > 128 unsupported by the "
                "CDNA4 kernels; using the composite fallback", d)
        return None
    return ext


def _pad_d(t: torch.Tensor, pad_to: int) -> torch.Tensor:
    return torch.nn.functional.pad(t, (0, pad_to - t.shape[-1]))


def _check_qkv(q, k, v):
    if q.is_cuda:
        assert q.dtype in (torch.float16, torch.bfloat16, torch.float32), \
            "flash attention requires fp16/bf16 (or fp32 composite)"
    assert q.dtype == k.dtype == v.dtype
    assert q.dim() == 4 and k.dim() == 4 and v.dim() == 4, \
        "expected [batch, seqlen, heads, head_dim]"
    assert k.shape == v.shape
    assert q.shape[0] == k.shape[0] and q.shape[3] == k.shape[3]
    assert q.shape[2] % k.shape[2] == 0, "GQA requires h_k | h_q"


def _ref_attention(q, k, v, softmax_scale, causal, window, q_lens=None,
                   k_lens=None, alibi_slopes=None):
    """fp32 composite reference; returns (out, lse[b,h,sq])."""
    b, sq, h, d = q.shape
    sk = k.shape[1]
    hk = k.shape[2]
    rep = h // hk
    qf = q.float().permute(0, 2, 1, 3)             # [b,h,sq,d]
    kf = k.float().permute(0, 2, 1, 3)             # [b,hk,sk,d]
    vf = v.float().permute(0, 2, 1, 3)
    if rep > 1:
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * softmax_scale
    neg = torch.finfo(torch.float32).min
    iq = torch.arange(sq, device=q.device).view(1, 1, sq, 1)
    ik = torch.arange(sk, device=q.device).view(1, 1, 1, sk)
    if alibi_slopes is not None:
        sl = alibi_slopes.float().view(1, h, 1, 1)
        scores = scores - sl * (ik - (iq + (sk - sq))).abs().float()
    # causal alignment: bottom-right (FA2 semantics)
    if causal:
        shift = sk - sq
        scores = scores.masked_fill(ik > iq + shift, neg)
    wl, wr = window
    if wl >= 0:
        shift = sk - sq
        scores = scores.masked_fill(ik < iq + shift - wl, neg)
    if wr >= 0 and not causal:
        shift = sk - sq
        scores = scores.masked_fill(ik > iq + shift + wr, neg)
    if k_lens is not None:
        klen = k_lens.view(b, 1, 1, 1)
        scores = scores.masked_fill(ik >= klen, neg)
    lse = torch.logsumexp(scores, dim=-1)          # [b,h,sq]
    p = torch.exp(scores - lse.unsqueeze(-1))
    p = torch.nan_to_num(p)                        # fully-masked rows
    out = torch.matmul(p, vf)                      # [b,h,sq,d]
    if q_lens is not None:
        qmask = (iq.view(1, 1, sq, 1) < q_lens.view(b, 1, 1, 1))
        out = out * qmask
        lse = torch.where(qmask.squeeze(-1), lse,
                          torch.zeros_like(lse))
    return out.permute(0, 2, 1, 3).to(q.dtype), lse


class FlashAttnFunc(torch.autograd.Function):

    @staticmethod
    def forward(ctx, q, k, v, dropout_p, softmax_scale, causal, window_size,
                alibi_slopes, deterministic, q_lens, k_lens):
        _check_qkv(q, k, v)
        if softmax_scale is None:
            softmax_scale = 1.0 / math.sqrt(q.shape[-1])
        q, k, v = [t.contiguous() for t in (q, k, v)]
        ext = _kernel_ext(q)
        wl, wr = window_size
        al = alibi_slopes if alibi_slopes is not None else torch.empty(0)
        cached = _sac_pop()
        if cached is not None:
            out, lse, seed = cached
        else:
            seed = 0
            if dropout_p > 0.0:
                assert ext is not None, \
                    "attention dropout runs only on the CDNA4 kernels (GPU)"
                # host-RNG seed: reproducible under torch.manual_seed; the
                # backward kernels regenerate the same keep-mask from it
                seed = int(torch.randint(0, 2 ** 62, (1,)).item())
            if ext is not None:
                d = q.shape[-1]
                pad_to = _pad_target(d)
                if pad_to:
                    # zero-padded head dims: QK^T and the valid output dims
                    # are unchanged; the pad columns of out are zero
                    out, lse = ext.fa_forward(
                        _pad_d(q, pad_to), _pad_d(k, pad_to),
                        _pad_d(v, pad_to), softmax_scale, causal, wl, wr,
                        q_lens if q_lens is not None else torch.empty(0),
                        k_lens if k_lens is not None else torch.empty(0),
                        al, dropout_p, seed)
                    out = out[..., :d].contiguous()
                else:
                    out, lse = ext.fa_forward(
                        q, k, v, softmax_scale, causal, wl, wr,
                        q_lens if q_lens is not None else torch.empty(0),
                        k_lens if k_lens is not None else torch.empty(0),
                        al, dropout_p, seed)
            else:
                out, lse = _ref_attention(q, k, v, softmax_scale, causal,
                                          (wl, wr), q_lens, k_lens,
                                          alibi_slopes)
        _sac_push((out.detach(), lse.detach(), seed))
        ctx.save_for_backward(
            q, k, v, out, lse,
            q_lens if q_lens is not None else torch.empty(0),
            k_lens if k_lens is not None else torch.empty(0), al)
        ctx.softmax_scale = softmax_scale
        ctx.causal = causal
        ctx.window = (wl, wr)
        ctx.deterministic = deterministic
        ctx.dropout = (dropout_p, seed)
        return out, lse

    @staticmethod
    def backward(ctx, dout, _dlse):
        q, k, v, out, lse, q_lens, k_lens, al = ctx.saved_tensors
        q_lens = q_lens if q_lens.numel() else None
        k_lens = k_lens if k_lens.numel() else None
        ext = _kernel_ext(q)
        dout = dout.contiguous()
        wl, wr = ctx.window
        if ext is not None:
            p_drop, seed = ctx.dropout
            d = q.shape[-1]
            pad_to = _pad_target(d)
            if pad_to:
                dq, dk, dv = ext.fa_backward(
                    _pad_d(dout, pad_to), _pad_d(q, pad_to),
                    _pad_d(k, pad_to), _pad_d(v, pad_to),
                    _pad_d(out, pad_to), lse, ctx.softmax_scale,
                    ctx.causal, wl, wr,
                    q_lens if q_lens is not None else torch.empty(0),
                    k_lens if k_lens is not None else torch.empty(0), al,
                    p_drop, seed)
                dq = dq[..., :d].contiguous()
                dk = dk[..., :d].contiguous()
                dv = dv[..., :d].contiguous()
            else:
                dq, dk, dv = ext.fa_backward(
                    dout, q, k, v, out, lse, ctx.softmax_scale, ctx.causal,
                    wl, wr,
                    q_lens if q_lens is not None else torch.empty(0),
                    k_lens if k_lens is not None else torch.empty(0), al,
                    p_drop, seed)
        else:
            dq, dk, dv = _ref_fa_backward(dout, q, k, v, out, lse,
                                          ctx.softmax_scale, ctx.causal,
                                          (wl, wr), q_lens, k_lens,
                                          al if al.numel() else None)
        return (dq, dk, dv) + (None,) * 8


def _ref_fa_backward(dout, q, k, v, out, lse, softmax_scale, causal, window,
                     q_lens, k_lens, alibi_slopes=None):
    """Recompute-based fp32 reference backward.

    ``out``/``lse`` are the GLOBAL attention output / logsumexp: delta =
    rowsum(dout * out) so the same routine serves both plain FA backward and
    per-block ring-attention backward (where p is normalized by the global
    lse and delta must come from the merged output)."""
    b, sq, h, d = q.shape
    sk, hk = k.shape[1], k.shape[2]
    rep = h // hk
    qf = q.float().permute(0, 2, 1, 3)
    kf = k.float().permute(0, 2, 1, 3)
    vf = v.float().permute(0, 2, 1, 3)
    dof = dout.float().permute(0, 2, 1, 3)
    if rep > 1:
        kf = kf.repeat_interleave(rep, dim=1)
        vf = vf.repeat_interleave(rep, dim=1)
    scores = torch.matmul(qf, kf.transpose(-1, -2)) * softmax_scale
    neg = torch.finfo(torch.float32).min
    iq = torch.arange(sq, device=q.device).view(1, 1, sq, 1)
    ik = torch.arange(sk, device=q.device).view(1, 1, 1, sk)
    shift = sk - sq
    if alibi_slopes is not None:
        sl = alibi_slopes.float().view(1, h, 1, 1)
        scores = scores - sl * (ik - (iq + shift)).abs().float()
    if causal:
        scores = scores.masked_fill(ik > iq + shift, neg)
    wl, wr = window
    if wl >= 0:
        scores = scores.masked_fill(ik < iq + shift - wl, neg)
    if wr >= 0 and not causal:
        scores = scores.masked_fill(ik > iq + shift + wr, neg)
    if k_lens is not None:
        scores = scores.masked_fill(ik >= k_lens.view(b, 1, 1, 1), neg)
    p = torch.exp(scores - lse.unsqueeze(-1))
    p = torch.nan_to_num(p)
    if q_lens is not None:
        qmask = iq < q_lens.view(b, 1, 1, 1)
        p = p * qmask
        dof = dof * qmask
    dv = torch.matmul(p.transpose(-1, -2), dof)
    dp = torch.matmul(dof, vf.transpose(-1, -2))
    delta = (dof * out.float().permute(0, 2, 1, 3)).sum(-1, keepdim=True)
    ds = p * (dp - delta) * softmax_scale
    dq = torch.matmul(ds, kf)
    dk = torch.matmul(ds.transpose(-1, -2), qf)
    if rep > 1:
        dk = dk.view(b, hk, rep, sk, d).sum(2)
        dv = dv.view(b, hk, rep, sk, d).sum(2)
    return (dq.permute(0, 2, 1, 3).to(q.dtype),
            dk.permute(0, 2, 1, 3).to(k.dtype),
            dv.permute(0, 2, 1, 3).to(v.dtype))


# ---------------------------------------------------------------------------
# user wrappers (reference-compatible names)
# ---------------------------------------------------------------------------

def _cast_fp16(q, k, v):
    """fp16 is now a NATIVE kernel dtype (v_mfma_f32_32x32x16_f16 via the
    AttnElem traits) — this shim is a passthrough kept for API stability
    (reference accepts fp16, ops/flash_attn.py:324-325)."""
    return q, k, v, False


def flash_attn_xla(q, k, v, dropout_p=0.0, softmax_scale=None, causal=False,
                   window_size=(-1, -1), alibi_slopes=None,
                   deterministic=False, return_attn_probs=False):
    """Fixed-length flash attention, q/k/v [b, s, h, d].

    Name kept from the reference for drop-in compatibility; runs on the
    CDNA4 HIP kernel (there is no XLA here).
    """
    q, k, v, back = _cast_fp16(q, k, v)
    out, lse = FlashAttnFunc.apply(q, k, v, dropout_p, softmax_scale, causal,
                                   window_size, alibi_slopes, deterministic,
                                   None, None)
    if back:
        out = out.to(torch.float16)
    if return_attn_probs:
        return out, lse, None
    return out


flash_attn_func = flash_attn_xla


def flash_attn_varlen_xla(q, k, v, attention_mask=None, dropout_p=0.0,
                          softmax_scale=None, causal=False,
                          window_size=(-1, -1), alibi_slopes=None,
                          deterministic=False, return_attn_probs=False):
    """Varlen-by-mask flash attention (reference FlashAttnVarlenXla,
    ops/flash_attn.py:219-262): ``attention_mask`` is an int mask [b, s_k]
    marking valid keys (right padding)."""
    k_lens = None
    q_lens = None
    if attention_mask is not None:
        assert attention_mask.dim() == 2
        k_lens = attention_mask.to(torch.int32).sum(-1).to(torch.int32)
        if q.shape[1] == k.shape[1]:
            q_lens = k_lens
    q, k, v, back = _cast_fp16(q, k, v)
    out, lse = FlashAttnFunc.apply(q, k, v, dropout_p, softmax_scale, causal,
                                   window_size, alibi_slopes, deterministic,
                                   q_lens, k_lens)
    if back:
        out = out.to(torch.float16)
    if return_attn_probs:
        return out, lse, None
    return out


def position_ids_to_cu_seqlens(position_ids: torch.Tensor) -> torch.Tensor:
    """[1, total] packed position ids -> int32 cu_seqlens (resets at 0),
    reference FlashAttnVarlenPositionIdsXla semantics."""
    pos = position_ids.reshape(-1)
    starts = torch.nonzero(pos == 0).reshape(-1)
    total = pos.numel()
    cu = torch.cat([starts.to(torch.int32),
                    torch.tensor([total], dtype=torch.int32,
                                 device=pos.device)])
    return cu


def flash_attn_varlen_position_ids_xla(q, k, v, position_ids, dropout_p=0.0,
                                       softmax_scale=None, causal=False,
                                       window_size=(-1, -1),
                                       alibi_slopes=None,
                                       deterministic=False,
                                       return_attn_probs=False):
    """Packed-sequence flash attention: bsz must be 1, sequences delimited by
    position-id resets (reference ops/flash_attn.py:173-218, 488)."""
    assert q.shape[0] == 1, "position-ids varlen requires batch size 1"
    cu = position_ids_to_cu_seqlens(position_ids)
    max_len = int((cu[1:] - cu[:-1]).max())
    q, k, v, back = _cast_fp16(q, k, v)
    out, lse = FlashAttnVarlenFunc.apply(
        q.squeeze(0), k.squeeze(0), v.squeeze(0), cu, cu, max_len, max_len,
        dropout_p, softmax_scale, causal, window_size, deterministic)
    out = out.unsqueeze(0)
    if back:
        out = out.to(torch.float16)
    if return_attn_probs:
        return out, lse, None
    return out


class FlashAttnVarlenFunc(torch.autograd.Function):
    """True packed varlen with explicit cu_seqlens (used by ring attention
    and the position-ids path). q/k/v are [total, h, d]."""

    @staticmethod
    def forward(ctx, q, k, v, cu_q, cu_k, max_q, max_k, dropout_p,
                softmax_scale, causal, window_size, deterministic):
        assert dropout_p == 0.0
        if softmax_scale is None:
            softmax_scale = 1.0 / math.sqrt(q.shape[-1])
        q, k, v = [t.contiguous() for t in (q, k, v)]
        ext = _kernel_ext(q)
        wl, wr = window_size
        cached = _sac_pop()
        if cached is not None:
            out, lse = cached[0], cached[1]
        elif ext is not None:
            out, lse = _varlen_fwd_gpu(ext, q, k, v, cu_q, cu_k,
                                       softmax_scale, causal, wl, wr)
        else:
            out, lse = _ref_varlen(q, k, v, cu_q, cu_k, softmax_scale,
                                   causal, (wl, wr))
        _sac_push((out.detach(), lse.detach(), 0))
        ctx.save_for_backward(q, k, v, out, lse, cu_q, cu_k)
        ctx.meta = (max_q, max_k, softmax_scale, causal, window_size)
        return out, lse

    @staticmethod
    def backward(ctx, dout, _dlse):
        q, k, v, out, lse, cu_q, cu_k = ctx.saved_tensors
        max_q, max_k, softmax_scale, causal, window_size = ctx.meta
        ext = _kernel_ext(q)
        dout = dout.contiguous()
        wl, wr = window_size
        if ext is not None:
            dq, dk, dv = _varlen_bwd_gpu(ext, dout, q, k, v, out, lse, cu_q,
                                         cu_k, softmax_scale, causal, wl, wr)
        else:
            dq, dk, dv = _ref_varlen_backward(dout, q, k, v, out, lse, cu_q,
                                              cu_k, softmax_scale, causal,
                                              (wl, wr))
        return (dq, dk, dv) + (None,) * 9


def _cu_to_bounds(cu: torch.Tensor, total: int) -> torch.Tensor:
    """cu_seqlens -> per-row int32 [total, 2] sequence intervals (the form
    the fused varlen kernels consume: same-sequence masking collapses to a
    per-q-row interval test, see csrc/flash_attn_varlen.hip)."""
    cu = cu.to(torch.int32)
    lens = (cu[1:] - cu[:-1]).long()
    starts = torch.repeat_interleave(cu[:-1], lens)
    ends = torch.repeat_interleave(cu[1:], lens)
    b = torch.stack([starts, ends], dim=1)
    if b.shape[0] < total:
        # rows past cu[-1] (padding): empty interval -> zero out/grads
        b = torch.cat([b, torch.zeros(total - b.shape[0], 2,
                                      dtype=torch.int32,
                                      device=b.device)])
    return b.contiguous()


def _use_fused_varlen(q, k, cu_q, cu_k, wl, wr):
    # `cu_q is cu_k` short-circuit: the position-ids and qkv-packed paths
    # pass the same tensor for both, avoiding a device sync (torch.equal)
    # on every attention call
    return (wl < 0 and wr < 0 and q.shape[0] == k.shape[0] and
            cu_q.numel() == cu_k.numel() and
            (cu_q is cu_k or bool(torch.equal(cu_q, cu_k))))


def _varlen_fwd_gpu(ext, q, k, v, cu_q, cu_k, softmax_scale, causal, wl,
                    wr):
    """Packed varlen: ONE fused kernel launch over the whole packed batch
    when q/k share the packing and there is no sliding window; otherwise
    one launch per sequence on the fixed-length kernel."""
    d = q.shape[-1]
    pad_to = _pad_target(d)
    if pad_to:
        out, lse = _varlen_fwd_gpu(ext, _pad_d(q, pad_to),
                                   _pad_d(k, pad_to), _pad_d(v, pad_to),
                                   cu_q, cu_k, softmax_scale, causal, wl,
                                   wr)
        return out[..., :d].contiguous(), lse
    total, h, _d = q.shape
    if _use_fused_varlen(q, k, cu_q, cu_k, wl, wr):
        bounds = _cu_to_bounds(cu_q.to(q.device), total)
        return tuple(ext.fa_varlen_forward(q, k, v, bounds, softmax_scale,
                                           causal))
    out = torch.zeros_like(q)
    lse = torch.zeros(h, total, dtype=torch.float32, device=q.device)
    empty = torch.empty(0, device=q.device)
    for i in range(cu_q.numel() - 1):
        qs, qe = int(cu_q[i]), int(cu_q[i + 1])
        ks, ke = int(cu_k[i]), int(cu_k[i + 1])
        if qe == qs:
            continue
        o_i, lse_i = ext.fa_forward(
            q[qs:qe].unsqueeze(0).contiguous(),
            k[ks:ke].unsqueeze(0).contiguous(),
            v[ks:ke].unsqueeze(0).contiguous(), softmax_scale, causal, wl,
            wr, empty, empty, empty, 0.0, 0)
        out[qs:qe] = o_i.squeeze(0)
        lse[:, qs:qe] = lse_i.squeeze(0)
    return out, lse


def _varlen_bwd_gpu(ext, dout, q, k, v, out, lse, cu_q, cu_k, softmax_scale,
                    causal, wl, wr):
    d = q.shape[-1]
    pad_to = _pad_target(d)
    if pad_to:
        dq, dk, dv = _varlen_bwd_gpu(
            ext, _pad_d(dout, pad_to), _pad_d(q, pad_to),
            _pad_d(k, pad_to), _pad_d(v, pad_to), _pad_d(out, pad_to),
            lse, cu_q, cu_k, softmax_scale, causal, wl, wr)
        return (dq[..., :d].contiguous(), dk[..., :d].contiguous(),
                dv[..., :d].contiguous())
    if _use_fused_varlen(q, k, cu_q, cu_k, wl, wr):
        bounds = _cu_to_bounds(cu_q.to(q.device), q.shape[0])
        return tuple(ext.fa_varlen_backward(dout, q, k, v, out, lse.contiguous(),
                                            bounds, softmax_scale, causal))
    dq = torch.zeros_like(q)
    dk = torch.zeros_like(k)
    dv = torch.zeros_like(v)
    empty = torch.empty(0, device=q.device)
    for i in range(cu_q.numel() - 1):
        qs, qe = int(cu_q[i]), int(cu_q[i + 1])
        ks, ke = int(cu_k[i]), int(cu_k[i + 1])
        if qe == qs:
            continue
        dq_i, dk_i, dv_i = ext.fa_backward(
            dout[qs:qe].unsqueeze(0).contiguous(),
            q[qs:qe].unsqueeze(0).contiguous(),
            k[ks:ke].unsqueeze(0).contiguous(),
            v[ks:ke].unsqueeze(0).contiguous(),
            out[qs:qe].unsqueeze(0).contiguous(),
            lse[:, qs:qe].unsqueeze(0).contiguous(), softmax_scale, causal,
            wl, wr, empty, empty, empty, 0.0, 0)
        dq[qs:qe] = dq_i.squeeze(0)
        dk[ks:ke] = dk_i.squeeze(0)
        dv[ks:ke] = dv_i.squeeze(0)
    return dq, dk, dv


def _ref_varlen(q, k, v, cu_q, cu_k, softmax_scale, causal, window):
    total, h, d = q.shape
    out = torch.zeros_like(q)
    lse = torch.full((h, total), 0.0, dtype=torch.float32, device=q.device)
    for i in range(cu_q.numel() - 1):
        qs, qe = int(cu_q[i]), int(cu_q[i + 1])
        ks, ke = int(cu_k[i]), int(cu_k[i + 1])
        if qe == qs:
            continue
        o, l = _ref_attention(q[qs:qe].unsqueeze(0), k[ks:ke].unsqueeze(0),
                              v[ks:ke].unsqueeze(0), softmax_scale, causal,
                              window)
        out[qs:qe] = o.squeeze(0)
        lse[:, qs:qe] = l.squeeze(0)
    return out, lse


def _ref_varlen_backward(dout, q, k, v, out, lse, cu_q, cu_k, softmax_scale,
                         causal, window):
    dq = torch.zeros_like(q)
    dk = torch.zeros_like(k)
    dv = torch.zeros_like(v)
    for i in range(cu_q.numel() - 1):
        qs, qe = int(cu_q[i]), int(cu_q[i + 1])
        ks, ke = int(cu_k[i]), int(cu_k[i + 1])
        if qe == qs:
            continue
        dqi, dki, dvi = _ref_fa_backward(
            dout[qs:qe].unsqueeze(0), q[qs:qe].unsqueeze(0),
            k[ks:ke].unsqueeze(0), v[ks:ke].unsqueeze(0),
            out[qs:qe].unsqueeze(0), lse[:, qs:qe].unsqueeze(0),
            softmax_scale, causal, window, None, None)
        dq[qs:qe] = dqi.squeeze(0)
        dk[ks:ke] = dki.squeeze(0)
        dv[ks:ke] = dvi.squeeze(0)
    return dq, dk, dv


def flash_attn_varlen_func(q, k, v, cu_seqlens_q, cu_seqlens_k, max_seqlen_q,
                           max_seqlen_k, dropout_p=0.0, softmax_scale=None,
                           causal=False, window_size=(-1, -1),
                           alibi_slopes=None, deterministic=False,
                           return_attn_probs=False):
    q, k, v, back = _cast_fp16(q, k, v)
    out, lse = FlashAttnVarlenFunc.apply(q, k, v, cu_seqlens_q, cu_seqlens_k,
                                         max_seqlen_q, max_seqlen_k,
                                         dropout_p, softmax_scale, causal,
                                         window_size, deterministic)
    if back:
        out = out.to(torch.float16)
    if return_attn_probs:
        return out, lse, None
    return out


# SPMD-named alias kept for API compatibility (reference ops/flash_attn.py:66)
def spmd_flash_attn_varlen_xla(*args, **kwargs):
    return flash_attn_varlen_xla(*args, **kwargs)


def flash_attn_qkvpacked_xla(qkv, dropout_p=0.0, softmax_scale=None,
                             causal=False, window_size=(-1, -1),
                             alibi_slopes=None, deterministic=False,
                             return_attn_probs=False):
    """qkv [b, s, 3, h, d] (reference FlashAttnVarlenQKVPackedXla:11)."""
    q, k, v = qkv.unbind(2)
    return flash_attn_xla(q, k, v, dropout_p, softmax_scale, causal,
                          window_size, alibi_slopes, deterministic,
                          return_attn_probs)


def flash_attn_varlen_qkvpacked_xla(qkv, attention_mask=None, **kw):
    q, k, v = qkv.unbind(2)
    return flash_attn_varlen_xla(q, k, v, attention_mask=attention_mask,
                                 **kw)


def flash_attn_kvpacked_xla(q, kv, dropout_p=0.0, softmax_scale=None,
                            causal=False, window_size=(-1, -1),
                            alibi_slopes=None, deterministic=False,
                            return_attn_probs=False):
    k, v = kv.unbind(2)
    return flash_attn_xla(q, k, v, dropout_p, softmax_scale, causal,
                          window_size, alibi_slopes, deterministic,
                          return_attn_probs)
