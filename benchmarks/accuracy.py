"""Convergence-parity harness (reference benchmarks/accuracy/run.sh:36-142):
train the same model twice — plain torch eager vs through
torchacc_amd.accelerate() — on identical data and assert the loss curves
agree to within the reference's 1e-2 gate.

Usage:  python benchmarks/accuracy.py [--steps 30] [--tol 1e-2]
        torchrun --nproc-per-node N ... benchmarks/accuracy.py  (FSDP=N)
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch  # noqa: E402


def make_model(seed=0):
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(seed)
    return LlamaForCausalLM(llama_tiny())


HF_VOCAB = 8192


def make_hf_model(seed=0, flash=False):
    """Small random-init HF transformers Llama (no network for real
    checkpoints; the reference's accuracy benchmark used Llama-3.2-1B —
    same architecture family, scaled down)."""
    from transformers.models.llama.configuration_llama import LlamaConfig
    from transformers.models.llama.modeling_llama import LlamaForCausalLM
    cfg = LlamaConfig(
        vocab_size=HF_VOCAB, hidden_size=512, intermediate_size=1024,
        num_hidden_layers=4, num_attention_heads=8, num_key_value_heads=8,
        max_position_embeddings=512,
        attn_implementation="flash_attention_2" if flash else "eager")
    torch.manual_seed(seed)
    return LlamaForCausalLM(cfg)


def data_stream(steps, seed=9, vocab=1024):
    torch.manual_seed(seed)
    return [torch.randint(0, vocab, (4, 64)) for _ in range(steps)]


def train_plain_hf(steps, device):
    """Pure-transformers baseline: must run BEFORE torchacc_amd is
    imported (kernel patches are class-level and import-time)."""
    assert "torchacc_amd" not in sys.modules, \
        "plain HF baseline must train before torchacc_amd patches load"
    model = make_hf_model().to(device)
    if device == "cuda":
        model = model.to(torch.bfloat16)
    opt = torch.optim.AdamW(model.parameters(), lr=3e-4, weight_decay=0.0)
    losses = []
    for ids in data_stream(steps, vocab=HF_VOCAB):
        ids = ids.to(device)
        loss = model(input_ids=ids, labels=ids).loss
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    return losses


def train_accelerated_hf(steps, world, device):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    cfg.compute.bf16 = device == "cuda"
    model = make_hf_model(flash=device == "cuda")
    model = ta.accelerate(model, config=cfg)
    opt = ta.ops.AdamW(model.parameters(), lr=3e-4, weight_decay=0.0)
    losses = []
    for ids in data_stream(steps, vocab=HF_VOCAB):
        ids = ids.to(ta.lazy_device())
        loss = model(input_ids=ids, labels=ids).loss
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    return losses


def train_plain(steps, device):
    model = make_model().to(device)
    opt = torch.optim.AdamW(model.parameters(), lr=3e-4, weight_decay=0.0)
    losses = []
    for ids in data_stream(steps):
        ids = ids.to(device)
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    return losses


def train_accelerated(steps, world):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    cfg.memory.gc = True
    cfg.memory.gc_cls = {"LlamaDecoderLayer"}
    model = make_model()
    model = ta.accelerate(model, config=cfg)
    opt = ta.ops.AdamW(model.parameters(), lr=3e-4, weight_decay=0.0)
    losses = []
    for ids in data_stream(steps):
        ids = ids.to(ta.lazy_device())
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    return losses


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--tol", type=float, default=1e-2)
    p.add_argument("--model", default="native", choices=["native",
                                                         "hf-llama"])
    args = p.parse_args()
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))
    device = "cuda" if torch.cuda.is_available() else "cpu"

    if args.model == "hf-llama":
        # order matters: the pure-HF baseline first (pre-patch)
        plain = train_plain_hf(args.steps, device)
        acc = train_accelerated_hf(args.steps, world, device)
    else:
        acc = train_accelerated(args.steps, world)
        plain = train_plain(args.steps, device)
    diff = max(abs(a - b) for a, b in zip(acc, plain))
    final_diff = abs(acc[-1] - plain[-1])
    if rank == 0:
        print(f"plain final {plain[-1]:.4f}  accelerated final "
              f"{acc[-1]:.4f}  max step diff {diff:.4f}  "
              f"final diff {final_diff:.4f}")
        tol = args.tol if device == "cpu" else max(args.tol, 5e-2)
        assert final_diff <= tol, \
            f"convergence parity FAILED: |{acc[-1]} - {plain[-1]}| > {tol}"
        print(f"CONVERGENCE PARITY OK (tol {tol})")


if __name__ == "__main__":
    main()
