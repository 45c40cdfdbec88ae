"""Tensor parallelism: 2-way TP matches single-process training."""
import pytest
import torch

from tests.utils.distributed import run_multiprocess


def _tp_worker(rank, world, q):
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    cfg = ta.Config()
    cfg.dist.tp.size = world
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    torch.manual_seed(42)
    losses = []
    for _ in range(4):
        ids = torch.randint(0, 1024, (2, 32))
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    q.put((rank, losses))


def test_tp2_matches_single():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_tp_worker, world_size=2, args=(q,))
    results = {}
    while not q.empty():
        r, losses = q.get()
        results[r] = losses
    assert len(results) == 2
    assert results[0] == pytest.approx(results[1], abs=1e-5)

    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    torch.manual_seed(42)
    base = []
    for _ in range(4):
        ids = torch.randint(0, 1024, (2, 32))
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        base.append(float(loss))
    assert results[0] == pytest.approx(base, abs=2e-4)


def _col_row_worker(rank, world, q):
    import torchacc_amd as ta
    import torch.distributed as dist
    from torchacc_amd.dist.tp import (ColumnParallelLinear,
                                      RowParallelLinear)
    group = dist.group.WORLD
    torch.manual_seed(0)
    lin1 = torch.nn.Linear(16, 32, bias=False)
    lin2 = torch.nn.Linear(32, 16, bias=False)
    col = ColumnParallelLinear.from_linear(lin1, group)
    row = RowParallelLinear.from_linear(lin2, group)
    torch.manual_seed(3)
    x = torch.randn(4, 16, requires_grad=True)
    y = row(col(x))
    ref = lin2(lin1(x))
    ok_fwd = torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    xg = x.grad.clone()
    x.grad = None
    x2 = x.detach().requires_grad_(True)
    lin2(lin1(x2)).backward(g)
    ok_bwd = torch.allclose(xg, x2.grad, atol=1e-5)
    q.put((rank, ok_fwd, ok_bwd))


def test_column_row_pair():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_col_row_worker, world_size=2, args=(q,))
    for _ in range(2):
        rank, ok_fwd, ok_bwd = q.get()
        assert ok_fwd, f"rank {rank} forward mismatch"
        assert ok_bwd, f"rank {rank} backward mismatch"


def _tp_fsdp_worker(rank, world, q):
    """tp2 x fsdp2 on 4 ranks: TP shards heads/MLP inside each FSDP
    replica group; losses must agree across all ranks and decrease."""
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    cfg = ta.Config()
    cfg.dist.tp.size = 2
    cfg.dist.fsdp.size = 2
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.SGD(model.parameters(), lr=0.05)
    torch.manual_seed(42)  # same data everywhere
    data = [torch.randint(0, 1024, (2, 32)) for _ in range(4)]
    losses = []
    for i in range(4):
        ids = data[i]
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    q.put((rank, losses))


def test_tp2_fsdp2_compose():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_tp_fsdp_worker, world_size=4, args=(q,))
    results = {}
    for _ in range(4):
        r, losses = q.get()
        results[r] = losses
    for r in range(1, 4):
        assert results[r] == pytest.approx(results[0], abs=1e-4), r
    assert results[0][-1] < results[0][0]
