"""Config-driven context parallelism through accelerate() (2-proc gloo):
sp.size=2 ulysses mode on the native Llama with cp_mode set."""
import numpy as np
import os
import torch

from tests.utils.distributed import run_multiprocess


def _cp_worker(rank, world, outdir):
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    cfg = ta.Config()
    cfg.dist.sp.size = world
    cfg.dist.sp.mode = "ulysses"
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny(cp_mode="ulysses"))
    model = ta.accelerate(model, config=cfg)
    torch.manual_seed(7)
    full_ids = torch.randint(0, 1024, (2, 64))
    # sequence sharded over cp ranks
    shard = full_ids[:, rank * 32:(rank + 1) * 32]
    loss = model(shard, labels=shard)
    loss.backward()
    np.save(os.path.join(outdir, f"loss_{rank}.npy"),
            np.array([float(loss)]))


def test_cp_ulysses_through_accelerate(tmp_path):
    run_multiprocess(_cp_worker, world_size=2, args=(str(tmp_path),))
    l0 = np.load(tmp_path / "loss_0.npy")[0]
    l1 = np.load(tmp_path / "loss_1.npy")[0]
    # each rank's loss is the mean CE over its sequence shard; with equal
    # shard sizes the average must match the single-process full-sequence
    # loss (ulysses attention sees the full sequence)
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    torch.manual_seed(7)
    full_ids = torch.randint(0, 1024, (2, 64))
    ref = float(model(full_ids, labels=full_ids))
    got = (l0 + l1) / 2
    assert abs(got - ref) < 5e-2, (got, ref)
