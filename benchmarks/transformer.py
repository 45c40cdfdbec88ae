"""Configurable transformer training-throughput harness
(reference benchmarks/transformer.py): any model size / strategy /
precision, optional torch.profiler trace, per-step samples/s + tokens/s.

Examples:
  python benchmarks/transformer.py --layers 4 --hidden 256 --steps 10
  torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
      benchmarks/transformer.py --mode fsdp --bf16 --gc --profile
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
    p.add_argument("--layers", type=int, default=4)
    p.add_argument("--hidden", type=int, default=512)
    p.add_argument("--heads", type=int, default=8)
    p.add_argument("--kv-heads", type=int, default=8)
    p.add_argument("--intermediate", type=int, default=1376)
    p.add_argument("--vocab", type=int, default=32000)
    p.add_argument("--seq-len", type=int, default=1024)
    p.add_argument("--batch-size", type=int, default=4)
    p.add_argument("--steps", type=int, default=10)
    p.add_argument("--warmup", type=int, default=3)
    p.add_argument("--mode", default="none",
                   choices=["none", "dp", "fsdp"])
    p.add_argument("--bf16", action="store_true")
    p.add_argument("--gc", action="store_true")
    p.add_argument("--profile", action="store_true",
                   help="emit a torch.profiler chrome trace")
    p.add_argument("--trace-dir", default="profiles/tb")
    args = p.parse_args()

    import torchacc_amd as ta
    from torchacc_amd.models import LlamaConfig, LlamaForCausalLM

    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    cfg = ta.Config()
    cfg.compute.bf16 = args.bf16 and torch.cuda.is_available()
    if args.mode == "dp":
        cfg.dist.dp.size = world
    elif args.mode == "fsdp":
        cfg.dist.fsdp.size = world
        cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    if args.gc:
        cfg.memory.gc = True
        cfg.memory.gc_cls = {"LlamaDecoderLayer"}

    torch.manual_seed(0)
    mcfg = LlamaConfig(
        vocab_size=args.vocab, hidden_size=args.hidden,
        intermediate_size=args.intermediate,
        num_hidden_layers=args.layers, num_attention_heads=args.heads,
        num_key_value_heads=args.kv_heads,
        max_position_embeddings=max(args.seq_len, 512))
    model = LlamaForCausalLM(mcfg)
    model = ta.accelerate(model, config=cfg)
    opt = ta.ops.AdamW(model.parameters(), lr=1e-4)
    device = ta.lazy_device()
    torch.manual_seed(1 + rank)
    ids = torch.randint(0, args.vocab, (args.batch_size, args.seq_len),
                        device=device)

    def step():
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)
        return loss

    for _ in range(args.warmup):
        step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()

    prof = None
    if args.profile:
        os.makedirs(args.trace_dir, exist_ok=True)
        prof = torch.profiler.profile(
            activities=[torch.profiler.ProfilerActivity.CPU,
                        torch.profiler.ProfilerActivity.CUDA],
            on_trace_ready=torch.profiler.tensorboard_trace_handler(
                args.trace_dir))
        prof.__enter__()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
        if prof is not None:
            prof.step()
    if torch.cuda.is_available():
        torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    if prof is not None:
        prof.__exit__(None, None, None)

    samples_s = args.batch_size * world * args.steps / dt
    tokens_s = samples_s * args.seq_len
    if rank == 0:
        print(json.dumps({
            "samples_per_sec": samples_s,
            "tokens_per_sec": tokens_s,
            "ms_per_step": dt / args.steps * 1000,
            "n_gpus": world,
            "mode": args.mode,
            "bf16": cfg.compute.bf16,
            "gc": args.gc,
        }))


if __name__ == "__main__":
    main()
