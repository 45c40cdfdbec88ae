"""Unit tests: async loader bucketing, cpu-offload context (CPU no-op path),
amp scaler, lazy-API shims, fused-kernel patching."""
import torch

import torchacc_amd as ta


def test_bucketing_loader_shapes():
    """reference tests/core/test_bucketing_loader.py:26-61"""
    from torchacc_amd.async_loader import (AsyncLoader, _get_closest_bucket,
                                           _uniform_buckets)
    assert _uniform_buckets(512, 4) == [128, 256, 384, 512]
    assert _get_closest_bucket([128, 256, 512], 100) == 128
    assert _get_closest_bucket([128, 256, 512], 300) == 512
    assert _get_closest_bucket([128, 256, 512], 600) == 512

    data = [{"input_ids": torch.ones(2, n, dtype=torch.long)}
            for n in (100, 200, 500)]
    loader = AsyncLoader(data, "cpu", buckets=[128, 256, 512])
    shapes = [b["input_ids"].shape[-1] for b in loader]
    assert shapes == [128, 256, 512]


def test_async_loader_prefetch_order():
    data = [{"x": torch.full((1, 4), i)} for i in range(10)]
    loader = ta.AsyncLoader(data, "cpu")
    vals = [int(b["x"][0, 0]) for b in loader]
    assert vals == list(range(10))
    assert len(loader) == 10


def test_cpu_offload_context_noop_on_cpu():
    from torchacc_amd.utils.cpu_offload import get_cpu_offload_context
    ctx, commit = get_cpu_offload_context(2)
    lin1 = torch.nn.Linear(8, 8)
    lin2 = torch.nn.Linear(8, 8)
    x = torch.randn(4, 8, requires_grad=True)
    with ctx:
        h = torch.relu(lin1(x))
    h = commit(h)
    with ctx:
        y = lin2(h)
    y = commit(y)
    y.sum().backward()
    assert x.grad is not None
    assert lin1.weight.grad is not None


def test_grad_scaler_cpu_disabled():
    scaler = ta.amp.GradScaler()
    lin = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(lin.parameters(), lr=0.1)
    x = torch.randn(2, 4)
    loss = lin(x).sum()
    scaler.scale(loss).backward()
    scaler.step(opt)
    scaler.update()
    assert lin.weight.grad is not None


def test_lazy_shims():
    assert ta.is_lazy_tensor(torch.ones(1)) is False
    ta.sync()
    ta.mark_step()
    t = torch.ones(3)
    assert ta.mark_dynamic(t) is t
    moved = ta.send_cpu_data_to_device({"a": t}, torch.device("cpu"))
    assert torch.equal(moved["a"], t)


def test_fetch_gradients():
    lin = torch.nn.Linear(4, 4)
    opt = torch.optim.SGD(lin.parameters(), lr=0.1)
    lin(torch.randn(2, 4)).sum().backward()
    grads = ta.fetch_gradients(opt)
    assert len(grads) == 2


def test_save_and_load(tmp_path):
    obj = {"w": torch.randn(3)}
    ta.save(obj, tmp_path / "x.pt")
    back = torch.load(tmp_path / "x.pt", weights_only=False)
    assert torch.equal(back["w"], obj["w"])


def test_patch_amp_swaps_adamw():
    from torchacc_amd.utils import patch
    orig = torch.optim.AdamW
    try:
        patch.patch_amp()
        from torchacc_amd.ops.adamw import AdamW
        assert torch.optim.AdamW is AdamW
    finally:
        torch.optim.AdamW = orig


def test_hf_kernel_patches_apply():
    try:
        import transformers  # noqa: F401
    except ImportError:
        return
    from torchacc_amd.utils import patch
    assert patch.apply_fused_kernel_patches() is True
    from transformers.models.llama import modeling_llama
    # patched RMSNorm runs our rms_norm
    m = modeling_llama.LlamaRMSNorm(16)
    x = torch.randn(2, 16)
    y = m(x)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) +
                          m.variance_epsilon) * m.weight
    assert torch.allclose(y, ref, atol=1e-5)


def test_qwen2_native_trains():
    from torchacc_amd.models import Qwen2ForCausalLM, qwen2_tiny
    torch.manual_seed(0)
    model = Qwen2ForCausalLM(qwen2_tiny(sliding_window=16))
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(1)
    ids = torch.randint(0, 1024, (2, 64))
    losses = []
    for _ in range(5):
        loss = model(ids, labels=ids)
        loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(loss))
    assert losses[-1] < losses[0], losses


def test_selective_attn_checkpoint_matches_full_recompute():
    """gc_selective_attn replays cached attention outputs during recompute;
    gradients must match plain checkpointing exactly."""
    import torch
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny

    def run(selective):
        torch.manual_seed(0)
        model = LlamaForCausalLM(llama_tiny())
        cfg = ta.Config()
        cfg.memory.gc = True
        cfg.memory.gc_cls = {"LlamaDecoderLayer"}
        cfg.memory.gc_selective_attn = selective
        model = ta.accelerate(model, config=cfg)
        torch.manual_seed(1)
        ids = torch.randint(0, 1024, (2, 64))
        loss = model(ids, labels=ids)
        loss.backward()
        grads = {n: p.grad.clone() for n, p in model.named_parameters()
                 if p.grad is not None}
        return float(loss), grads

    l_full, g_full = run(False)
    l_sel, g_sel = run(True)
    assert l_full == l_sel
    assert g_full.keys() == g_sel.keys() and len(g_full) > 0
    for n in g_full:
        assert torch.equal(g_full[n], g_sel[n]), n


def test_llama_gqa_forward_backward():
    """GQA (kv heads < heads) end-to-end on the composite CPU path; the
    llama-3-8b factory exposes the same config at scale."""
    import torch
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny, llama_3_8b
    cfg3 = llama_3_8b()
    assert cfg3.num_key_value_heads == 8 and cfg3.rope_theta == 500000.0
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny(num_key_value_heads=2))
    ids = torch.randint(0, 1024, (2, 32))
    loss = model(ids, labels=ids)
    loss.backward()
    assert torch.isfinite(loss)


def test_patch_qwen_model_rebinds_remote_flash_attn():
    """patch_qwen_model points a remote-code module's flash-attn symbols at
    this framework's ops (reference regex-rewrote source; we rebind)."""
    import sys
    import types
    import torch
    from torchacc_amd import patch_qwen_model
    from torchacc_amd.ops.flash_attn import flash_attn_varlen_func

    fake = types.ModuleType("remote_qwen_modeling")
    fake.flash_attn_unpadded_func = lambda *a, **k: None
    sys.modules["remote_qwen_modeling"] = fake
    try:
        class FakeQwen(torch.nn.Module):
            pass
        FakeQwen.__module__ = "remote_qwen_modeling"
        m = FakeQwen()
        assert patch_qwen_model(m) is True
        assert fake.flash_attn_unpadded_func is flash_attn_varlen_func
    finally:
        del sys.modules["remote_qwen_modeling"]


def test_bucketing_pad_value_dict():
    """pad_value_dict controls the fill per key (labels -100 so CE ignores
    the padding)."""
    from torchacc_amd.async_loader import AsyncLoader
    data = [{"input_ids": torch.ones(2, 100, dtype=torch.long),
             "labels": torch.ones(2, 100, dtype=torch.long)}]
    loader = AsyncLoader(data, "cpu", buckets=[128],
                         pad_value_dict={"labels": -100})
    batch = next(iter(loader))
    assert batch["input_ids"].shape[-1] == 128
    assert (batch["input_ids"][:, 100:] == 0).all()
    assert (batch["labels"][:, 100:] == -100).all()


def test_mem_plan_tool():
    import subprocess
    import sys
    import os
    root = os.path.dirname(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))))
    out = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "mem_plan.py"),
         "--model", "llama-2-70b", "--bs", "2", "--seq", "4096",
         "--fsdp", "8"],
        capture_output=True, text=True, check=True)
    assert "recommended gc_cnt" in out.stdout
