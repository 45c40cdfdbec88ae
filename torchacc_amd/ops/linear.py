"""Projection GEMMs on hipBLASLt with pinned (offline-tuned) algorithms.

PyTorch's matmul takes hipBLASLt's heuristic top-1 — measured ~52-60% of
bf16 peak on the Llama training shapes (profiles/r01_7b_fsdp1.md, 67% of
step time). ``benchmarks/gemm_tune.py`` times the full solution space per
shape on the target GPU and writes ``gemm_algos_gfx950.json``; this module
routes the three linear-layer GEMMs (fwd y=xW^T, dgrad dx=dyW, wgrad
dW=dy^T x) through ``_C.lt_gemm`` with the cached winner. Shapes without a
tuned entry use hipBLASLt's heuristic through the same code path, so
behavior is uniform and never silently eager.

Replaces the reference's reliance on XLA's gemm autotuner (the lazy
backend's cublas algo search happened inside XLA; eager MI355X needs it
done by hand).
"""
import functools
import json
import os
from typing import Optional

import torch
import torch.nn.functional as F

from ._backend import _load

_ALGO_FILE = os.path.join(os.path.dirname(os.path.abspath(__file__)),
                          "gemm_algos_gfx950.json")


@functools.lru_cache(maxsize=1)
def _algo_table() -> dict:
    if not os.path.exists(_ALGO_FILE):
        return {}
    try:
        with open(_ALGO_FILE) as f:
            data = json.load(f)
        return {k: int(v["algo"]) for k, v in data.get("entries", {}).items()}
    except (json.JSONDecodeError, KeyError, TypeError):
        return {}


def _algo_for(m: int, n: int, k: int, ta: bool, tb: bool) -> int:
    if os.environ.get("TA_GEMM_HEURISTIC") == "1":
        return -1  # A/B escape hatch: heuristic top-1 everywhere
    key = f"{m},{n},{k},{'t' if ta else 'n'}{'t' if tb else 'n'}"
    return _algo_table().get(key, -1)


class _LtLinear(torch.autograd.Function):
    """y = x @ W^T on lt_gemm, with dgrad/dwgrad on their tuned algos."""

    @staticmethod
    def forward(ctx, x, weight, ext):
        ctx.save_for_backward(x, weight)
        ctx.ext = ext
        m, k = x.shape
        n = weight.shape[0]
        return ext.lt_gemm(x, weight, False, True, _algo_for(m, n, k, False,
                                                             True))

    @staticmethod
    def backward(ctx, dy):
        x, weight = ctx.saved_tensors
        ext = ctx.ext
        dy = dy.contiguous()
        m, k = x.shape
        n = weight.shape[0]
        dx = dw = None
        if ctx.needs_input_grad[0]:
            dx = ext.lt_gemm(dy, weight, False, False,
                             _algo_for(m, k, n, False, False))
        if ctx.needs_input_grad[1]:
            dw = ext.lt_gemm(dy, x, True, False,
                             _algo_for(n, k, m, True, False))
        return dx, dw, None


def tuned_linear(x: torch.Tensor, weight: torch.Tensor,
                 bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear drop-in; hipBLASLt with pinned algos on GPU bf16."""
    ext = None if os.environ.get("TA_DISABLE_TUNED_GEMM") == "1" else _load()
    if (ext is None or not x.is_cuda or bias is not None
            or x.dtype not in (torch.bfloat16, torch.float16)
            or weight.dtype != x.dtype):
        return F.linear(x, weight, bias)
    shape = x.shape
    x2d = x.reshape(-1, shape[-1])
    if not x2d.is_contiguous():
        x2d = x2d.contiguous()
    if x2d.shape[0] <= 4 and not torch.is_grad_enabled() \
            and x2d.shape[-1] % 8 == 0:
        # decode-shaped: the wave-per-row GEMV streams W once and beats
        # hipBLASLt's skinny kernels at m<=4 (b1 decode 217->225 tok/s,
        # +38% under hipGraph); at m=8 the per-wave x reloads lose to
        # hipBLASLt, so larger micro-batches take the lt path
        out = ext.lt_gemv(x2d, weight)
    elif x.dtype == torch.bfloat16:
        out = _LtLinear.apply(x2d, weight, ext)
    else:
        return F.linear(x, weight, bias)
    return out.view(*shape[:-1], weight.shape[0])


class TunedLinear(torch.nn.Linear):
    """nn.Linear whose forward runs on the tuned hipBLASLt path."""

    def forward(self, x):
        return tuned_linear(x, self.weight, self.bias)
