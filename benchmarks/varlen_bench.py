"""Fused varlen attention vs per-sequence kernel loop: wall time for one
packed batch forward+backward (the fused path is one launch per kernel
instead of one per sequence)."""
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
    p.add_argument("--nseq", type=int, default=16)
    p.add_argument("--seqlen", type=int, default=512)
    p.add_argument("--h", type=int, default=32)
    p.add_argument("--hk", type=int, default=8)
    p.add_argument("--d", type=int, default=128)
    args = p.parse_args()
    from torchacc_amd.ops._backend import require_extension
    from torchacc_amd.ops.flash_attn import _cu_to_bounds
    ext = require_extension()
    torch.manual_seed(0)
    lens = [args.seqlen] * args.nseq
    total = sum(lens)
    h, hk, d = args.h, args.hk, args.d
    cu = torch.tensor([0] + list(torch.cumsum(torch.tensor(lens), 0)),
                      dtype=torch.int32, device="cuda")
    q = torch.randn(total, h, d, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(total, hk, d, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(total, hk, d, device="cuda", dtype=torch.bfloat16)
    do = torch.randn_like(q)
    scale = d ** -0.5
    bounds = _cu_to_bounds(cu, total)
    e = torch.empty(0, device="cuda")

    def loop_fwd():
        out = torch.zeros_like(q)
        lse = torch.zeros(h, total, dtype=torch.float32, device="cuda")
        for i in range(cu.numel() - 1):
            qs, qe = i * args.seqlen, (i + 1) * args.seqlen
            o_i, l_i = ext.fa_forward(
                q[qs:qe].unsqueeze(0).contiguous(),
                k[qs:qe].unsqueeze(0).contiguous(),
                v[qs:qe].unsqueeze(0).contiguous(), scale, True, -1, -1,
                e, e, e, 0.0, 0)
            out[qs:qe] = o_i.squeeze(0)
            lse[:, qs:qe] = l_i.squeeze(0)
        return out, lse

    out, lse = ext.fa_varlen_forward(q, k, v, bounds, scale, True)
    t_loop_f = bench(loop_fwd)
    t_fused_f = bench(
        lambda: ext.fa_varlen_forward(q, k, v, bounds, scale, True))
    t_fused_b = bench(
        lambda: ext.fa_varlen_backward(do, q, k, v, out, lse, bounds, scale,
                                       True))
    print(json.dumps({
        "shape": f"nseq{args.nseq} s{args.seqlen} h{h} hk{hk} d{d} causal",
        "loop_fwd_ms": t_loop_f * 1e3,
        "fused_fwd_ms": t_fused_f * 1e3,
        "fused_bwd_ms": t_fused_b * 1e3,
        "fwd_speedup_vs_loop": t_loop_f / t_fused_f,
    }))


if __name__ == "__main__":
    main()
