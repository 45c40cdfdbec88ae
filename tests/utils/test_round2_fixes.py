"""Round-2 correctness fixes (advisor findings + syncfree AMP contract).

- gradient-checkpoint wrappers must not leak '_checkpoint_wrapped_module.'
  into state dicts or FSDP shard metadata (gc-on and gc-off checkpoints are
  interchangeable, matching the reference checkpoint format);
- micro-batch splitting requires divisibility (uneven chunks would be
  over-weighted by the executor's 1/num scaling);
- the HF MLP kernel patch honors config.hidden_act;
- ops.AdamW implements torch's _step_supports_amp_scaling contract
  (device-side grad_scale/found_inf; reference syncfree semantics,
  utils/patch.py:55-57).
"""
import pytest
import torch

from tests.utils.distributed import run_multiprocess


def test_checkpoint_wrapper_state_dict_prefix():
    from torchacc_amd.utils.checkpoint import gradient_checkpoint

    torch.manual_seed(0)

    def make():
        return torch.nn.Sequential(
            torch.nn.Linear(8, 8), torch.nn.Linear(8, 8))

    plain = make()
    wrapped = make()
    wrapped.load_state_dict(plain.state_dict())
    gradient_checkpoint(wrapped, gc_cls={"Linear"})
    sd = wrapped.state_dict()
    assert set(sd.keys()) == set(plain.state_dict().keys())
    for k in sd:
        assert "_checkpoint_wrapped_module" not in k
    # a clean (gc-off) state dict loads into the gc-wrapped model
    wrapped.load_state_dict(plain.state_dict())
    # and the gc-wrapped state dict loads into a clean model
    plain.load_state_dict(sd)


def _gc_shard_names_worker(rank, world):
    import torchacc_amd as ta
    from torchacc_amd.models import LlamaForCausalLM, llama_tiny
    cfg = ta.Config()
    cfg.dist.fsdp.size = world
    cfg.dist.fsdp.wrap_layer_cls = {"LlamaDecoderLayer"}
    cfg.memory.gc = True
    cfg.memory.gc_cls = {"LlamaDecoderLayer"}
    torch.manual_seed(0)
    model = LlamaForCausalLM(llama_tiny())
    clean_keys = {k for k in model.state_dict().keys()
                  if not k.startswith("rope_")}
    wrapped = ta.accelerate(model, config=cfg)
    meta = wrapped.fsdp_wrapper.shard_metadata()
    names = {p["name"] for u in meta["units"] for p in u["params"]}
    for n in names:
        assert "_checkpoint_wrapped_module" not in n, n
    assert names == clean_keys, names ^ clean_keys
    full = wrapped.full_state_dict()
    for k in full:
        assert "_checkpoint_wrapped_module" not in k, k


def test_fsdp_gc_shard_metadata_names_clean():
    run_multiprocess(_gc_shard_names_worker, world_size=2)


def test_microbatch_divisibility_required():
    from torchacc_amd.dist.pp.microbatch import split_microbatches
    ok = split_microbatches({"x": torch.zeros(8, 3)}, 4)
    assert len(ok) == 4 and all(m["x"].shape[0] == 2 for m in ok)
    with pytest.raises(AssertionError):
        split_microbatches({"x": torch.zeros(6, 3)}, 4)


def test_hf_mlp_patch_honors_hidden_act():
    transformers = pytest.importorskip("transformers")
    from transformers.models.llama.configuration_llama import LlamaConfig
    from transformers.models.llama.modeling_llama import LlamaMLP

    from torchacc_amd.utils.patch import apply_fused_kernel_patches

    apply_fused_kernel_patches()
    torch.manual_seed(0)
    x = torch.randn(2, 4, 32)

    gelu_cfg = LlamaConfig(hidden_size=32, intermediate_size=64,
                           hidden_act="gelu")
    mlp = LlamaMLP(gelu_cfg)
    got = mlp(x)
    want = mlp.down_proj(
        torch.nn.functional.gelu(mlp.gate_proj(x)) * mlp.up_proj(x))
    assert torch.allclose(got, want, atol=1e-6), \
        "gelu MLP must not be silently replaced by silu"

    silu_cfg = LlamaConfig(hidden_size=32, intermediate_size=64,
                           hidden_act="silu")
    mlp2 = LlamaMLP(silu_cfg)
    got2 = mlp2(x)
    want2 = mlp2.down_proj(
        torch.nn.functional.silu(mlp2.gate_proj(x)) * mlp2.up_proj(x))
    assert torch.allclose(got2, want2, atol=1e-5)


def test_adamw_step_supports_amp_scaling():
    from torchacc_amd.ops.adamw import AdamW

    assert AdamW._step_supports_amp_scaling is True

    torch.manual_seed(0)
    p_ref = torch.nn.Parameter(torch.randn(16))
    p_scaled = torch.nn.Parameter(p_ref.detach().clone())
    g = torch.randn(16)

    opt_ref = AdamW([p_ref], lr=1e-2)
    p_ref.grad = g.clone()
    opt_ref.step()

    # grads still carry the loss scale; the GradScaler contract hands the
    # optimizer grad_scale/found_inf attributes and the step unscales itself
    scale = 1024.0
    opt = AdamW([p_scaled], lr=1e-2)
    p_scaled.grad = g * scale
    opt.grad_scale = torch.tensor(scale)
    opt.found_inf = torch.zeros(())
    opt.step()
    assert torch.allclose(p_scaled.detach(), p_ref.detach(), atol=1e-6)


def test_adamw_skips_on_found_inf():
    from torchacc_amd.ops.adamw import AdamW
    p = torch.nn.Parameter(torch.randn(8))
    before = p.detach().clone()
    opt = AdamW([p], lr=1.0)
    p.grad = torch.randn(8)
    opt.step(found_inf=torch.ones(()))
    assert torch.equal(p.detach(), before)


def test_hf_lce_forward_loss_parity_and_no_logits():
    """HF causal-LM forward routes lm_head+loss through chunked fused
    linear-CE (Liger lce_forward parity, reference ops/liger.py:75-76):
    loss matches the unpatched model, logits are never materialized."""
    pytest.importorskip("transformers")
    from transformers.models.llama.configuration_llama import LlamaConfig
    from transformers.models.llama.modeling_llama import LlamaForCausalLM

    from torchacc_amd.utils.patch import (_ORIG_CAUSAL_FWD,
                                          apply_fused_kernel_patches)

    apply_fused_kernel_patches()
    assert LlamaForCausalLM in _ORIG_CAUSAL_FWD
    cfg = LlamaConfig(vocab_size=128, hidden_size=32, intermediate_size=64,
                      num_hidden_layers=2, num_attention_heads=4,
                      num_key_value_heads=4, max_position_embeddings=64,
                      attn_implementation="eager")
    torch.manual_seed(0)
    model = LlamaForCausalLM(cfg)
    ids = torch.randint(0, 128, (2, 16))
    out = model(input_ids=ids, labels=ids)
    assert out.logits is None, "fused path must not materialize logits"
    ref = _ORIG_CAUSAL_FWD[LlamaForCausalLM](model, input_ids=ids,
                                             labels=ids)
    assert abs(float(out.loss) - float(ref.loss)) < 1e-4
    # inference path (no labels) untouched: full logits
    infer = model(input_ids=ids)
    assert infer.logits is not None and infer.logits.shape[-1] == 128
    # grad-accumulation normalization contract (HF num_items_in_batch)
    out2 = model(input_ids=ids, labels=ids, num_items_in_batch=60)
    want = float(ref.loss) * 2 * 15 / 60
    assert abs(float(out2.loss) - want) < 1e-4


def test_hf_rope_patch_matches_eager():
    pytest.importorskip("transformers")
    from transformers.models.llama.modeling_llama import rotate_half

    from torchacc_amd.utils.patch import _hf_rope_forward

    torch.manual_seed(0)
    b, h, s, d = 2, 4, 16, 32
    q = torch.randn(b, h, s, d)
    k = torch.randn(b, 2, s, d)
    half = torch.randn(1, s, d // 2)
    cos = torch.cat([half.cos(), half.cos()], dim=-1)
    sin = torch.cat([half.sin(), half.sin()], dim=-1)
    qo, ko = _hf_rope_forward(q, k, cos, sin)
    c, si = cos.unsqueeze(1), sin.unsqueeze(1)
    q_ref = q * c + rotate_half(q) * si
    k_ref = k * c + rotate_half(k) * si
    assert torch.allclose(qo, q_ref, atol=1e-5)
    assert torch.allclose(ko, k_ref, atol=1e-5)
