"""Flash-attention kernel microbenchmark: TFLOP/s fwd and bwd on random
data (guide §5.4 rule 25: never bench attention on zero-filled data)."""
import argparse
import json
import sys
import time
import os

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--b", type=int, default=16)
    p.add_argument("--s", type=int, default=2048)
    p.add_argument("--h", type=int, default=64)
    p.add_argument("--hk", type=int, default=8)
    p.add_argument("--d", type=int, default=128)
    p.add_argument("--causal", action="store_true")
    args = p.parse_args()
    from torchacc_amd.ops._backend import require_extension
    ext = require_extension()
    torch.manual_seed(0)
    b, s, h, hk, d = args.b, args.s, args.h, args.hk, args.d
    q = torch.randn(b, s, h, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(b, s, hk, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(b, s, hk, d, device="cuda", dtype=torch.bfloat16)
    do = torch.randn_like(q)
    scale = d ** -0.5
    causal = args.causal
    flops_fwd = 4 * b * h * s * s * d * (0.5 if causal else 1.0)

    e = torch.empty(0)
    out, lse = ext.fa_forward(q, k, v, scale, causal, -1, -1, e, e, e,
                              0.0, 0)
    t_fwd = bench(lambda: ext.fa_forward(q, k, v, scale, causal, -1, -1,
                                         e, e, e, 0.0, 0))
    t_bwd = bench(lambda: ext.fa_backward(do, q, k, v, out, lse, scale,
                                          causal, -1, -1, e, e, e, 0.0, 0))
    print(json.dumps({
        "shape": f"b{b} s{s} h{h} hk{hk} d{d} causal={causal}",
        "fwd_ms": t_fwd * 1e3,
        "fwd_tflops": flops_fwd / t_fwd / 1e12,
        "bwd_ms": t_bwd * 1e3,
        "bwd_tflops": 2.5 * flops_fwd / t_bwd / 1e12,
    }))


if __name__ == "__main__":
    main()
