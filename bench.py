"""Flagship benchmark: Llama-2-7B FSDP bf16 training throughput (tokens/s,
whole job) on MI355X — the BASELINE.json headline metric.

Launch (single GPU):  python bench.py --steps 10 --warmup 3
Launch (N GPUs):      python -m torch.distributed.run --nnodes=1
                      --nproc-per-node N --master-addr 127.0.0.1 bench.py
                      --gpus N --steps K --warmup W

Synthetic data (random token ids of the target shape), random-init weights.
Timing: W untimed warmup steps, then exactly K steps bracketed by
barrier + torch.cuda.synchronize() on both sides; MAX step time over ranks;
rank 0 prints ONE JSON line.
"""
import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--model", default="llama-2-7b",
                   choices=["llama-2-7b", "llama-2-70b", "llama-3-8b",
                            "qwen2-7b", "tiny"])
    p.add_argument("--seq-len", type=int, default=4096)
    p.add_argument("--batch-size", type=int, default=8,
                   help="micro batch per GPU")
    p.add_argument("--mode", default="fsdp",
                   choices=["fsdp", "ulysses", "ring", "2d", "fsdp_tp"])
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--no-gc-selective", action="store_true",
                   help="full recompute inside checkpointed layers instead "
                        "of retaining attention outputs (selective AC)")
    p.add_argument("--no-gc", action="store_true",
                   help="disable gradient checkpointing")
    p.add_argument("--fp16", action="store_true",
                   help="train in fp16 (native f16 kernels + GradScaler)")
    p.add_argument("--gc-cnt", type=int, default=None,
                   help="checkpoint only the first N layers (288 GB HBM3E "
                        "rarely needs all of them)")
    return p.parse_args()


def build_model(args, cfg):
    from torchacc_amd.models import (LlamaConfig, LlamaForCausalLM,
                                     llama_2_70b, llama_2_7b, llama_3_8b,
                                     llama_tiny)
    cp_mode = args.mode if args.mode in ("ulysses", "ring", "2d") else None
    kw = dict(max_position_embeddings=max(args.seq_len, 4096),
              cp_mode=cp_mode)
    if args.model == "llama-2-7b":
        mcfg = llama_2_7b(**kw)
    elif args.model == "llama-2-70b":
        mcfg = llama_2_70b(**kw)
    elif args.model == "llama-3-8b":
        mcfg = llama_3_8b(**kw)
    elif args.model == "qwen2-7b":
        from torchacc_amd.models import Qwen2ForCausalLM, qwen2_7b
        assert cp_mode is None, "qwen2 bench supports fsdp modes"
        mcfg = qwen2_7b(
            max_position_embeddings=max(args.seq_len, 4096))
        torch.manual_seed(1234)
        return Qwen2ForCausalLM(mcfg), mcfg
    else:
        mcfg = llama_tiny(cp_mode=cp_mode)
    torch.manual_seed(1234)
    return LlamaForCausalLM(mcfg), mcfg


def main():
    args = parse_args()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    on_gpu = torch.cuda.is_available()

    import torchacc_amd as ta

    cfg = ta.Config()
    cfg.compute.bf16 = on_gpu and not args.fp16
    cfg.compute.fp16 = on_gpu and args.fp16
    parallelism = f"fsdp{world}"
    if args.mode == "fsdp":
        cfg.dist.fsdp.size = world
        cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer", "Qwen2DecoderLayer"}
    elif args.mode in ("ulysses", "ring", "2d"):
        cfg.dist.sp.size = world
        if args.mode == "ulysses":
            cfg.dist.sp.intra_size = world
        elif args.mode == "ring":
            cfg.dist.sp.intra_size = 1
        else:
            cfg.dist.sp.intra_size = max(1, world // 2)
        cfg.dist.sp.mode = args.mode
        parallelism = f"{args.mode}{world}"
    elif args.mode == "fsdp_tp":
        cfg.dist.tp.size = args.tp
        cfg.dist.fsdp.size = world // args.tp
        cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer", "Qwen2DecoderLayer"}
        parallelism = f"fsdp{world // args.tp}xtp{args.tp}"
    if not args.no_gc and args.model != "tiny":
        cfg.memory.gc = True
        cfg.memory.gc_cls = {"LlamaDecoderLayer", "Qwen2DecoderLayer"}
        cfg.memory.gc_selective_attn = not args.no_gc_selective
        gc_cnt = args.gc_cnt
        if gc_cnt is None and args.mode == "fsdp" \
                and args.batch_size * args.seq_len <= 32768:
            # 288 GB HBM3E rarely needs every layer checkpointed; depth
            # picked per model from same-box sweeps (profiles/r02):
            # llama-2-7b 20.6k at gc4 (206 GB) vs 16.6-17.0k at all-32;
            # llama-3-8b gc4 18.4k > gc8 18.0k (236 GB); qwen2-7b is the
            # exception — gc4 is 10% SLOWER than gc8 (17.1k vs 19.0k at
            # 229 GB), so it keeps 8 checkpointed layers.
            gc_cnt = {"llama-2-7b": 4, "llama-3-8b": 4,
                      "qwen2-7b": 8}.get(args.model)
        cfg.memory.gc_cnt = gc_cnt

    if on_gpu:
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", 0)))
        with torch.device("cuda"):
            model, mcfg = build_model(args, cfg)
    else:
        model, mcfg = build_model(args, cfg)
    model = ta.accelerate(model, config=cfg)
    opt = ta.ops.AdamW(model.parameters(), lr=1e-4, weight_decay=0.0)
    scaler = ta.amp.GradScaler() if cfg.compute.fp16 else None

    device = ta.lazy_device()
    seq = args.seq_len
    if args.model == "tiny":
        seq = min(seq, 512)  # tiny config's rope table / max_position
    bs = args.batch_size
    is_cp = args.mode in ("ulysses", "ring", "2d")
    local_seq = seq // world if is_cp else seq
    torch.manual_seed(5678 + rank if not is_cp else 5678)
    # a few distinct synthetic batches, cycled: keeps the loss honest (no
    # single-batch memorization) without growing device memory
    batches = [
        torch.randint(0, mcfg.vocab_size, (bs, local_seq), device=device)
        for _ in range(4)
    ]
    it = [0]

    def step_fixed():
        ids = batches[it[0] % len(batches)]
        it[0] += 1
        # labels are shifted inside the model (predict t+1 from t)
        loss = model(ids, labels=ids)
        if scaler is not None:
            scaler.scale(loss).backward()
            scaler.step(opt)
            scaler.update()
        else:
            loss.backward()
            opt.step()
        opt.zero_grad(set_to_none=True)
        return loss

    if dist.is_initialized() and world > 1:
        dist.barrier()
    for _ in range(args.warmup):
        step_fixed()
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized() and world > 1:
        dist.barrier()
    t0 = time.perf_counter()
    last = None
    for _ in range(args.steps):
        last = step_fixed()
    if on_gpu:
        torch.cuda.synchronize()
    if dist.is_initialized() and world > 1:
        dist.barrier()
    t1 = time.perf_counter()

    elapsed = torch.tensor([t1 - t0], dtype=torch.float64)
    if dist.is_initialized() and world > 1:
        # MAX over ranks
        e = elapsed.to(device if on_gpu else "cpu")
        dist.all_reduce(e, op=dist.ReduceOp.MAX)
        elapsed = e.cpu()
    secs = float(elapsed[0])
    ms_per_step = secs / args.steps * 1000.0
    if is_cp:
        tokens_per_step = bs * seq  # whole-job: sequence sharded over ranks
    else:
        tokens_per_step = bs * local_seq * world
    toks_per_s = tokens_per_step * args.steps / secs

    if rank == 0:
        out = {
            "metric": "tokens/sec",
            "value": toks_per_s,
            "unit": "tokens/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong" if is_cp else "weak",
            "vs_baseline": None,
            "dtype": ("fp16" if args.fp16 else "bf16") if on_gpu
                     else "fp32",
            "data": "synthetic",
            "config": {
                "model": args.model,
                "global_batch": bs * (1 if is_cp else world),
                "seq_len": seq,
                "parallelism": parallelism,
                "grad_checkpoint": ("selective_attn"
                                    if cfg.memory.gc and
                                    cfg.memory.gc_selective_attn
                                    else cfg.memory.gc),
                "gc_cnt": cfg.memory.gc_cnt,
                "loss": float(last.detach()) if last is not None else None,
                "peak_mem_gb": (round(
                    torch.cuda.max_memory_allocated() / 2**30, 2)
                    if on_gpu else None),
            },
        }
        print(json.dumps(out))


if __name__ == "__main__":
    main()
