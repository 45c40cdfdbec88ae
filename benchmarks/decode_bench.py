"""Serving-side decode throughput: KV-cache generate() on the native
models (prefill + steady-state single-token decode).

  python benchmarks/decode_bench.py --model llama-2-7b --batch 1 \
      --prompt 128 --new 128
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="llama-2-7b",
                   choices=["llama-2-7b", "llama-3-8b", "tiny"])
    p.add_argument("--batch", type=int, default=1)
    p.add_argument("--prompt", type=int, default=128)
    p.add_argument("--new", type=int, default=128)
    p.add_argument("--warmup", type=int, default=8)
    p.add_argument("--graph", action="store_true",
                   help="hipGraph-captured decode step (greedy)")
    args = p.parse_args()

    from torchacc_amd.models import (LlamaForCausalLM, llama_2_7b,
                                     llama_3_8b, llama_tiny)
    cfgs = {"llama-2-7b": llama_2_7b, "llama-3-8b": llama_3_8b,
            "tiny": llama_tiny}
    mcfg = cfgs[args.model](
        max_position_embeddings=max(4096, args.prompt + args.new + 1))
    torch.manual_seed(0)
    on_gpu = torch.cuda.is_available()
    dev = "cuda" if on_gpu else "cpu"
    if on_gpu:
        with torch.device(dev):
            model = LlamaForCausalLM(mcfg).to(torch.bfloat16).eval()
    else:
        model = LlamaForCausalLM(mcfg).eval()
    ids = torch.randint(0, mcfg.vocab_size, (args.batch, args.prompt),
                        device=dev)

    if args.graph:
        from torchacc_amd.models.generation import GraphDecoder
        dec = GraphDecoder(model, args.batch,
                           args.prompt + args.new + args.warmup + 2)
        dec.decode(ids, args.warmup)  # warm + capture
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = dec.decode(ids, args.new)
    else:
        model.generate(ids, max_new_tokens=args.warmup)  # warm
        if on_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        out = model.generate(ids, max_new_tokens=args.new)
    if on_gpu:
        torch.cuda.synchronize()
    t1 = time.perf_counter()
    new_tokens = (out.shape[1] - args.prompt) * args.batch
    print(json.dumps({
        "metric": "decode tokens/sec",
        "value": new_tokens / (t1 - t0),
        "ms_per_token": (t1 - t0) / (out.shape[1] - args.prompt) * 1000,
        "model": args.model, "batch": args.batch,
        "prompt": args.prompt, "new": args.new,
        "dtype": "bf16" if on_gpu else "fp32",
        "graph": args.graph,
    }))


if __name__ == "__main__":
    main()
