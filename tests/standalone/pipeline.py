"""Standalone PP smoke: 4-stage pipeline with gradient checkpointing on
the stage layers (reference tests/standalone/pipeline.py ran 4 stages).

torchrun --nproc-per-node 4 --master-addr 127.0.0.1 \
    tests/standalone/pipeline.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch  # noqa: E402

import torchacc_amd as ta  # noqa: E402
from torchacc_amd.models import LlamaForCausalLM, llama_tiny  # noqa: E402


def main():
    world = int(os.environ.get("WORLD_SIZE", 2))
    rank = int(os.environ.get("RANK", 0))
    cfg = ta.Config()
    cfg.dist.pp.size = world
    cfg.dist.pp.num_micro_batches = 4
    cfg.dist.pp.input_names = ["input_ids", "labels"]
    cfg.memory.gc = True
    cfg.memory.gc_cls = {"LlamaDecoderLayer"}
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    model = ta.accelerate(model, config=cfg)
    opt = ta.ops.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(11)
    ids = torch.randint(0, 1024, (8, 32))  # fixed batch (memorization)
    losses = []
    for _ in range(6):
        loss = model.forward_backward(ids, labels=ids)
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses
    if rank == 0:
        print(f"pipeline OK: {losses[0]:.3f} -> {losses[-1]:.3f}")


if __name__ == "__main__":
    main()
