"""Standalone accelerate() matrix (reference tests/standalone/ta_accelerate.py):
exercises DP / FSDP / FSDP+GC / ckpt save+consolidate+reshard round trip.

torchrun --nproc-per-node 2 --master-addr 127.0.0.1 \
    tests/standalone/ta_accelerate.py
"""
import os
import sys
import tempfile

sys.path.insert(0, os.path.join(os.path.dirname(__file__), "..", ".."))

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402

import torchacc_amd as ta  # noqa: E402
from torchacc_amd.models import LlamaForCausalLM, llama_tiny  # noqa: E402


def run_case(name, cfg_fn, steps=3):
    rank = int(os.environ.get("RANK", 0))
    cfg = ta.Config()
    cfg_fn(cfg)
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    model = ta.accelerate(model, config=cfg)
    opt = ta.ops.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(7)
    for _ in range(steps):
        ids = torch.randint(0, 1024, (2, 32))
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
    if rank == 0:
        print(f"  {name}: final loss {float(loss):.4f}")
    return model, opt


def main():
    world = int(os.environ.get("WORLD_SIZE", 1))
    rank = int(os.environ.get("RANK", 0))

    def dp_cfg(c):
        c.dist.dp.size = world

    def fsdp_cfg(c):
        c.dist.fsdp.size = world
        c.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}

    def fsdp_gc_cfg(c):
        fsdp_cfg(c)
        c.memory.gc = True
        c.memory.gc_cls = {"LlamaDecoderLayer"}

    run_case("dp", dp_cfg)
    run_case("fsdp+gc", fsdp_gc_cfg)
    model, opt = run_case("fsdp", fsdp_cfg)

    # checkpoint round trip
    from torchacc_amd.dist.state_dict_utils import (
        consolidate_and_reshard_fsdp_checkpoint, save_sharded_checkpoint)
    tmp = None
    if rank == 0:
        tmp = tempfile.mkdtemp()
    obj = [tmp]
    if dist.is_initialized() and world > 1:
        dist.broadcast_object_list(obj, src=0)
    tmp = obj[0]
    save_sharded_checkpoint(model, opt, tmp)
    if dist.is_initialized() and world > 1:
        dist.barrier()
    if rank == 0:
        consolidate_and_reshard_fsdp_checkpoint(tmp, tmp, reshard_num=4)
        assert os.path.exists(os.path.join(tmp, "consolidated_model.pth"))
        assert os.path.exists(os.path.join(tmp, "rank-3-of-4-model.pth"))
        print("  ckpt consolidate+reshard OK")
    if dist.is_initialized() and world > 1:
        dist.barrier()
    if rank == 0:
        print("ta_accelerate OK")


if __name__ == "__main__":
    main()
