"""Per-GPU memory plan for a model/parallelism configuration.

Sizes the training state the way the framework lays it out on 288 GB
HBM3E (bf16 flat-param FSDP shards + fused-AdamW fp32 states/masters +
activations under selective-AC with ``gc_cnt``), and recommends the
smallest gc_cnt that fits with headroom — partial-layer checkpointing is
the main 288-GB lever (profiles/r02_7b_fsdp1.md: gc 8/32 layers is +11%
over all-32 on Llama-2-7B).

Usage:
  python tools/mem_plan.py --model llama-2-7b --bs 8 --seq 4096 --fsdp 8
  python tools/mem_plan.py --model llama-2-70b --bs 2 --seq 4096 \
      --fsdp 1 --tp 8
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

HBM_GB = 288
GB = 1 << 30

MODELS = {
    # hidden, inter, layers, heads, kv_heads, vocab
    "llama-2-7b": (4096, 11008, 32, 32, 32, 32000),
    "llama-2-70b": (8192, 28672, 80, 64, 8, 32000),
    "llama-3-8b": (4096, 14336, 32, 32, 8, 128256),
    "llama-3-70b": (8192, 28672, 80, 64, 8, 128256),
    "qwen2-7b": (3584, 18944, 28, 28, 4, 152064),
}


def plan(model, bs, seq, fsdp, tp, dp=1, selective_attn=True):
    h, inter, L, nh, nkv, vocab = MODELS[model]
    d = h // nh
    layer_params = (h * (nh + 2 * nkv) * d + nh * d * h) + 3 * h * inter \
        + 2 * h
    embed_params = vocab * h
    head_params = vocab * h
    norm_params = h
    total_params = L * layer_params + embed_params + head_params + \
        norm_params

    shard = fsdp * tp  # params divided across fsdp shards AND tp slices
    # bf16 param shard + bf16 grad shard + fp32 (master, exp_avg,
    # exp_avg_sq) in the fused AdamW
    state_bytes = total_params / shard * (2 + 2 + 4 * 3)
    # one gathered unit (largest layer) resident during compute
    gathered_bytes = layer_params / tp * 2 * (2 if fsdp > 1 else 0)

    tokens = bs * seq
    # activations per NON-checkpointed layer (bf16): attn in/q/k/v/softmax
    # out/proj out + mlp in/gate/up/swiglu/down + 2 norms (+residuals)
    act_full = tokens * (
        h * 6 + (nh + 2 * nkv) * d + inter * 3) * 2 / tp
    # checkpointed layer keeps: layer input + (selective-AC) attention
    # out + lse
    act_ckpt = tokens * (h + (nh * d + nh / d if selective_attn else 0)) * 2
    # logits path: chunked linear-CE peak (8192-row chunk, fp32 internals)
    ce_bytes = min(tokens, 8192) * vocab / tp * (2 + 4)
    embed_bytes = tokens * h * 2

    def fits(gc_cnt):
        acts = (L - gc_cnt) * act_full + gc_cnt * act_ckpt
        return state_bytes + gathered_bytes + acts + ce_bytes + \
            embed_bytes + 8 * GB  # allocator/fragmentation reserve

    rec = None
    for gc in range(0, L + 1):
        if fits(gc) <= HBM_GB * GB * 0.92:
            rec = gc
            break
    return {
        "params_B": total_params / 1e9,
        "state_gb": state_bytes / GB,
        "act_full_layer_gb": act_full / GB,
        "act_ckpt_layer_gb": act_ckpt / GB,
        "ce_gb": ce_bytes / GB,
        "total_no_gc_gb": fits(0) / GB,
        "total_full_gc_gb": fits(L) / GB,
        "recommended_gc_cnt": rec,
        "total_at_rec_gb": fits(rec) / GB if rec is not None else None,
    }


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-7b", choices=sorted(MODELS))
    p.add_argument("--bs", type=int, default=8)
    p.add_argument("--seq", type=int, default=4096)
    p.add_argument("--fsdp", type=int, default=1)
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--full-recompute", action="store_true")
    args = p.parse_args()
    r = plan(args.model, args.bs, args.seq, args.fsdp, args.tp,
             selective_attn=not args.full_recompute)
    print(f"{args.model} bs{args.bs} seq{args.seq} fsdp{args.fsdp} "
          f"tp{args.tp}  ({r['params_B']:.1f}B params)")
    print(f"  states (param+grad+opt shards): {r['state_gb']:8.1f} GB")
    print(f"  activations/layer full:         {r['act_full_layer_gb']:8.2f}"
          " GB")
    print(f"  activations/layer checkpointed: {r['act_ckpt_layer_gb']:8.2f}"
          " GB")
    print(f"  peak, no checkpointing:         {r['total_no_gc_gb']:8.1f} GB")
    print(f"  peak, all layers checkpointed:  {r['total_full_gc_gb']:8.1f}"
          " GB")
    if r["recommended_gc_cnt"] is None:
        print("  DOES NOT FIT in 288 GB even fully checkpointed — raise "
              "fsdp/tp or lower bs/seq")
    else:
        print(f"  recommended gc_cnt: {r['recommended_gc_cnt']} "
              f"(peak ≈ {r['total_at_rec_gb']:.1f} GB of {HBM_GB})")


if __name__ == "__main__":
    main()
