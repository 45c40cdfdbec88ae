"""Offline hipBLASLt algorithm search for the training GEMM shapes.

Times every supported hipBLASLt solution (``_C.lt_gemm_candidates``) for the
fwd/dgrad/wgrad GEMMs of a model config — plus the chunked
linear-cross-entropy GEMMs — and writes the winners to
``torchacc_amd/ops/gemm_algos_gfx950.json`` (consumed by ops/linear.py).
Two-pass search: a quick 2-iteration sweep over the full space, then a
10-iteration refine of the top 12.

Run ON the GPU box:
  python benchmarks/gemm_tune.py --model llama-2-7b --bs 8 --seq 4096
  python benchmarks/gemm_tune.py --model llama-2-70b --bs 2 --seq 4096 --tp 8

The tuned-vs-heuristic delta per shape is printed so a flat result is
documented (profiles/) rather than silently shipped.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: E402

ALGO_FILE = os.path.join(
    os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
    "torchacc_amd", "ops", "gemm_algos_gfx950.json")

MODELS = {
    # hidden, intermediate, heads, kv_heads, vocab
    "llama-2-7b": (4096, 11008, 32, 32, 32000),
    "llama-2-70b": (8192, 28672, 64, 8, 32000),
    "llama-3-8b": (4096, 14336, 32, 8, 128256),
    "qwen2-7b": (3584, 18944, 28, 4, 152064),
}

CE_CHUNK = 8192  # ops/cross_entropy.py _LinearCrossEntropy.CHUNK


def shapes_for(model: str, bs: int, seq: int, tp: int):
    h, inter, heads, kv, vocab = MODELS[model]
    d = h // heads
    m = bs * seq
    qkv_n = (heads + 2 * kv) * d // tp
    o_n = h
    o_k = heads * d // tp
    gu_n = 2 * inter // tp
    dn_k = inter // tp
    shapes = set()

    def linear(mm, n, k):
        shapes.add((mm, n, k, "nt"))   # fwd  y = x W^T
        shapes.add((mm, k, n, "nn"))   # dgrad dx = dy W
        shapes.add((n, k, mm, "tn"))   # wgrad dW = dy^T x

    linear(m, qkv_n, h)                # fused qkv (also covers split: below)
    linear(m, heads * d // tp, h)      # q_proj (unfused path)
    if kv != heads:
        linear(m, kv * d // tp, h)     # k/v_proj (GQA, unfused path)
    linear(m, o_n, o_k)                # o_proj
    linear(m, gu_n, h)                 # fused gate_up
    linear(m, inter // tp, h)          # gate/up (unfused path)
    linear(m, h, dn_k)                 # down_proj
    # chunked linear-CE: logits chunk fwd (x2), dgrad, wgrad
    linear(CE_CHUNK, vocab // tp, h)
    return sorted(shapes)


ROT = 4  # rotate input copies so repeats never hit warm caches


def _mk(m, n, k, op, dev):
    ta, tb = op[0] == "t", op[1] == "t"
    mk = lambda shape: [  # noqa: E731
        torch.randn(shape, device=dev, dtype=torch.bfloat16)
        for _ in range(ROT)
    ]
    a = mk((k, m) if ta else (m, k))
    b = mk((n, k) if tb else (k, n))
    return a, b, ta, tb


def _time_algo(ext, a, b, ta, tb, algo, iters, out):
    """Median of per-iteration times over rotating cold inputs (a single
    hot-loop average rewards cache-resident algos that lose in context)."""
    ev = [torch.cuda.Event(True) for _ in range(iters + 1)]
    ext.lt_gemm(a[0], b[0], ta, tb, algo, out)  # warm/validate
    torch.cuda.synchronize()
    ev[0].record()
    for i in range(iters):
        ext.lt_gemm(a[(i + 1) % ROT], b[(i + 1) % ROT], ta, tb, algo, out)
        ev[i + 1].record()
    torch.cuda.synchronize()
    times = sorted(ev[i].elapsed_time(ev[i + 1]) for i in range(iters))
    return times[len(times) // 2]


def tune_shape(ext, m, n, k, op, dev, cap, budget_s=30.0):
    a, b, ta, tb = _mk(m, n, k, op, dev)
    out = torch.empty(m, n, device=dev, dtype=torch.bfloat16)
    flops = 2.0 * m * n * k
    base_ms = _time_algo(ext, a, b, ta, tb, -1, 10, out)
    cands = ext.lt_gemm_candidates(m, n, k, ta, tb, 256)
    if cap and len(cands) > cap:
        cands = cands[:cap]
    t0 = time.time()
    quick = []
    for idx in cands:
        if time.time() - t0 > budget_s:
            break
        try:
            ms = _time_algo(ext, a, b, ta, tb, idx, 2, out)
        except RuntimeError:
            continue
        quick.append((ms, idx))
    quick.sort()
    best_ms, best_idx = base_ms, -1
    for ms, idx in quick[:12]:
        ms10 = _time_algo(ext, a, b, ta, tb, idx, 10, out)
        if ms10 < best_ms:
            best_ms, best_idx = ms10, idx
    return {
        "algo": best_idx,
        "ms": round(best_ms, 4),
        "tflops": round(flops / best_ms / 1e9, 1),
        "base_ms": round(base_ms, 4),
        "base_tflops": round(flops / base_ms / 1e9, 1),
        "gain_pct": round((base_ms / best_ms - 1) * 100, 1),
        "n_cands": len(cands),
        "name": ext.lt_gemm_algo_name(best_idx) if best_idx >= 0 else "",
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-7b", choices=sorted(MODELS))
    p.add_argument("--bs", type=int, default=8)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--cap", type=int, default=2000,
                   help="max candidates per shape in the quick pass")
    p.add_argument("--budget", type=float, default=30.0,
                   help="seconds per shape for the quick pass")
    p.add_argument("--out", default=ALGO_FILE)
    args = p.parse_args()

    assert torch.cuda.is_available(), "run on the GPU box"
    from torchacc_amd.ops._backend import require_extension
    ext = require_extension()
    dev = torch.device("cuda")

    data = {"entries": {}}
    if os.path.exists(args.out):
        with open(args.out) as f:
            data = json.load(f)
    entries = data.setdefault("entries", {})

    shapes = shapes_for(args.model, args.bs, args.seq, args.tp)
    print(f"{len(shapes)} shapes for {args.model} bs{args.bs} seq{args.seq} "
          f"tp{args.tp}")
    for m, n, k, op in shapes:
        r = tune_shape(ext, m, n, k, op, dev, args.cap, args.budget)
        key = f"{m},{n},{k},{op}"
        entries[key] = r
        print(f"{key:>28}  base {r['base_tflops']:7.1f} TF -> "
              f"{r['tflops']:7.1f} TF  (+{r['gain_pct']}%)  algo "
              f"{r['algo']}  [{r['n_cands']} cands]")
        with open(args.out, "w") as f:
            json.dump(data, f, indent=1)
    print(f"wrote {args.out}")


if __name__ == "__main__":
    main()
