"""FSDP engine tests: 2-process gloo world, parity vs single-process."""
import pytest
import torch

from tests.utils.distributed import run_multiprocess


def _make_model(seed=0):
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    torch.manual_seed(seed)
    return LlamaForCausalLM(llama_tiny())


def _train_steps(model, opt, data, steps):
    losses = []
    for i in range(steps):
        ids = data[i]
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    return losses


def _fsdp_worker(rank, world, q_losses):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    model = _make_model()
    model = ta.accelerate(model, config=cfg)
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(42)  # same data on both ranks -> dp grad == local grad
    data = [torch.randint(0, 1024, (2, 32)) for _ in range(4)]
    losses = _train_steps(model, opt, data, 4)
    q_losses.put((rank, losses))


def test_fsdp2_matches_single_process():
    """2-way FSDP on identical data must match single-process training."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_fsdp_worker, world_size=2, args=(q,))
    results = {}
    while not q.empty():
        r, losses = q.get()
        results[r] = losses
    assert len(results) == 2
    assert results[0] == pytest.approx(results[1], abs=1e-5)

    # single-process baseline
    model = _make_model()
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(42)
    data = [torch.randint(0, 1024, (2, 32)) for _ in range(4)]
    base = _train_steps(model, opt, data, 4)
    assert results[0] == pytest.approx(base, abs=5e-3)


def _state_dict_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    model = _make_model()
    ref_sd = {k: v.clone() for k, v in model.state_dict().items()}
    wrapped = ta.accelerate(model, config=cfg)
    full = wrapped.full_state_dict()
    ok = True
    for k, v in ref_sd.items():
        if k.startswith("rope_"):
            continue
        if k not in full or not torch.allclose(full[k], v, atol=1e-6):
            ok = False
    q.put((rank, ok, sorted(full.keys())[:3]))


def test_fsdp_full_state_dict_roundtrip():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_state_dict_worker, world_size=2, args=(q,))
    for _ in range(2):
        rank, ok, sample = q.get()
        assert ok, f"rank {rank} full_state_dict mismatch ({sample})"


def _gc_reshard_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    cfg.memory.gc = True
    cfg.memory.gc_cls = {"LlamaDecoderLayer"}
    cfg.memory.gc_selective_attn = True  # SAC under FSDP regather
    model = _make_model()
    wrapped = ta.accelerate(model, config=cfg)
    fsdp = wrapped.fsdp_wrapper
    torch.manual_seed(3)
    ids = torch.randint(0, 1024, (2, 32))
    for _ in range(2):
        loss = wrapped(ids, labels=ids)
        loss.backward()
    # after backward every non-root unit must be resharded (pending == 0)
    bad = [u.name for u in fsdp.units
           if u is not fsdp.root_unit and (u.unsharded or u.pending_bwd)]
    q.put((rank, bad))


def test_gc_units_reshard_after_backward():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_gc_reshard_worker, world_size=2, args=(q,))
    for _ in range(2):
        rank, bad = q.get()
        assert not bad, f"rank {rank}: units left gathered {bad}"


def _no_sync_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    model = _make_model()
    wrapped = ta.accelerate(model, config=cfg)
    fsdp = wrapped.fsdp_wrapper
    torch.manual_seed(42)
    a = torch.randint(0, 1024, (2, 32))
    b = torch.randint(0, 1024, (2, 32))
    # accumulate two micro-steps under no_sync, reduce on the third
    with fsdp.no_sync():
        wrapped(a, labels=a).backward()
    wrapped(b, labels=b).backward()
    g1 = {i: u.shard.grad.clone() for i, u in enumerate(fsdp.units)}
    for u in fsdp.units:
        u.shard.grad = None
    # reference: two separate synchronized backwards (grads add)
    wrapped(a, labels=a).backward()
    wrapped(b, labels=b).backward()
    ok = all(
        torch.allclose(g1[i], u.shard.grad, atol=1e-5)
        for i, u in enumerate(fsdp.units))
    q.put((rank, ok))


def test_no_sync_grad_accumulation():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_no_sync_worker, world_size=2, args=(q,))
    for _ in range(2):
        rank, ok = q.get()
        assert ok, f"rank {rank} no_sync accumulation mismatch"


def _clip_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    model = _make_model()
    wrapped = ta.accelerate(model, config=cfg)
    torch.manual_seed(42)
    ids = torch.randint(0, 1024, (2, 32))
    wrapped(ids, labels=ids).backward()
    total = wrapped.clip_grad_norm_(0.1)
    # after clipping, recompute the global shard norm: must equal max_norm
    fsdp = wrapped.fsdp_wrapper
    import torch.distributed as dist
    local = sum(u.shard.grad.float().pow(2).sum() for u in fsdp.units
                if u.shard.grad is not None)
    dist.all_reduce(local)
    q.put((rank, float(total), float(local.sqrt())))


def test_clip_grad_norm_matches_single_process():
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_clip_worker, world_size=2, args=(q,))
    results = [q.get() for _ in range(2)]
    # pre-clip norm identical across ranks and equals the single-process one
    assert results[0][1] == pytest.approx(results[1][1], rel=1e-6)
    model = _make_model()
    torch.manual_seed(42)
    ids = torch.randint(0, 1024, (2, 32))
    model(ids, labels=ids).backward()
    ref = torch.nn.utils.clip_grad_norm_(model.parameters(), 0.1)
    assert results[0][1] == pytest.approx(float(ref), rel=1e-3)
    # post-clip global norm == max_norm
    for _, _, post in results:
        assert post == pytest.approx(0.1, rel=1e-3)


def _sync_states_worker(rank, world, q):
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    cfg.dist.fsdp.sync_module_states = True
    torch.manual_seed(rank * 97 + 1)  # DIFFERENT init per rank
    model = _make_model(seed=rank * 97 + 1)
    wrapped = ta.accelerate(model, config=cfg)
    full = wrapped.full_state_dict()
    sig = float(sum(v.float().sum() for v in full.values()))
    q.put((rank, sig))


def test_sync_module_states_broadcasts_rank0():
    """sync_module_states=True must make all ranks' full state identical
    (broadcast from the group source) despite different local inits."""
    import multiprocessing as mp
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    run_multiprocess(_sync_states_worker, world_size=2, args=(q,))
    sigs = {}
    for _ in range(2):
        r, s = q.get()
        sigs[r] = s
    assert sigs[0] == pytest.approx(sigs[1], rel=1e-6), sigs
