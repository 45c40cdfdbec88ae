"""HF-transformers integration: the patch layer routes a real HF Llama
through the framework's ops (the reference's primary user path,
benchmarks/accuracy/run_clm.py:59-62)."""
import pytest
import torch

transformers = pytest.importorskip("transformers")


def _tiny_hf_llama():
    from transformers import LlamaConfig, LlamaForCausalLM
    cfg = LlamaConfig(
        vocab_size=512, hidden_size=128, intermediate_size=344,
        num_hidden_layers=2, num_attention_heads=4, num_key_value_heads=4,
        max_position_embeddings=256,
        attn_implementation="eager")
    torch.manual_seed(0)
    return LlamaForCausalLM(cfg)


def test_patched_hf_llama_trains():
    import torchacc_amd as ta
    ta.accelerate_hf_trainer()  # applies fa + fused-kernel patches
    model = _tiny_hf_llama()
    opt = torch.optim.AdamW(model.parameters(), lr=1e-3)
    torch.manual_seed(1)
    ids = torch.randint(0, 512, (2, 64))
    losses = []
    for _ in range(5):
        out = model(input_ids=ids, labels=ids)
        out.loss.backward()
        opt.step()
        opt.zero_grad()
        losses.append(float(out.loss))
    assert losses[-1] < losses[0], losses


def test_patched_flash_attention_entry():
    """patch_fa routes transformers' _flash_attention_forward to our op and
    the output matches SDPA on the same inputs."""
    from torchacc_amd.utils import patch
    assert patch.patch_fa() is True
    from transformers import modeling_flash_attention_utils as mfa
    torch.manual_seed(0)
    b, s, h, d = 1, 32, 2, 16
    q = torch.randn(b, s, h, d)
    k = torch.randn(b, s, h, d)
    v = torch.randn(b, s, h, d)
    out = mfa._flash_attention_forward(q, k, v, None, s, is_causal=True)
    ref = torch.nn.functional.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        is_causal=True).transpose(1, 2)
    assert torch.allclose(out, ref, atol=1e-4)


def test_accelerate_wraps_hf_model_fsdp():
    """accelerate() + FSDP wrapping on an HF model (wrap_layer_cls matches
    HF's LlamaDecoderLayer)."""
    import torchacc_amd as ta
    cfg = ta.Config()
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    model = _tiny_hf_llama()
    wrapped = ta.accelerate(model, config=cfg)
    opt = torch.optim.AdamW(wrapped.parameters(), lr=1e-3)
    ids = torch.randint(0, 512, (2, 32))
    out = wrapped(input_ids=ids, labels=ids)
    out.loss.backward()
    opt.step()
    assert torch.isfinite(out.loss)


def test_hf_flash_attention_2_construction():
    """attn_implementation='flash_attention_2' must work WITHOUT the CUDA
    flash_attn package (our CDNA4 kernels are the implementation), and the
    runtime path must route through our patched entry point."""
    from torchacc_amd.utils import patch
    assert patch.patch_fa() is True
    import transformers.integrations.flash_attention as fa_int
    calls = {"n": 0}
    orig = fa_int._flash_attention_forward

    def counted(*a, **k):
        calls["n"] += 1
        return orig(*a, **k)

    fa_int._flash_attention_forward = counted
    try:
        from transformers import LlamaConfig, LlamaForCausalLM
        cfg = LlamaConfig(
            vocab_size=64, hidden_size=64, intermediate_size=128,
            num_hidden_layers=2, num_attention_heads=4,
            num_key_value_heads=4,
            attn_implementation="flash_attention_2")
        torch.manual_seed(0)
        m = LlamaForCausalLM(cfg)
        ids = torch.randint(0, 64, (2, 16))
        out = m(input_ids=ids, labels=ids)
        out.loss.backward()
        assert torch.isfinite(out.loss)
    finally:
        fa_int._flash_attention_forward = orig
    assert calls["n"] == 2, calls
