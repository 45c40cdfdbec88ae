"""CPU reference-path op tests: gradcheck-style parity of the composite
reference implementations (the same math the GPU numerics tests compare the
HIP kernels against)."""
import math

import pytest
import torch

from torchacc_amd.ops.cross_entropy import cross_entropy, linear_cross_entropy
from torchacc_amd.ops.flash_attn import (flash_attn_varlen_xla,
                                         flash_attn_xla)
from torchacc_amd.ops.rmsnorm import rms_norm
from torchacc_amd.ops.rope import apply_rotary_pos_emb, build_rope_cache
from torchacc_amd.ops.swiglu import swiglu


def test_rmsnorm_matches_composite():
    torch.manual_seed(0)
    x = torch.randn(4, 16, 64, requires_grad=True)
    w = torch.randn(64, requires_grad=True)
    y = rms_norm(x, w, 1e-6)
    ref = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    assert torch.allclose(y, ref, atol=1e-5)
    g = torch.randn_like(y)
    y.backward(g)
    xg, wg = x.grad.clone(), w.grad.clone()
    x.grad = None
    w.grad = None
    ref2 = x * torch.rsqrt(x.pow(2).mean(-1, keepdim=True) + 1e-6) * w
    ref2.backward(g)
    assert torch.allclose(xg, x.grad, atol=1e-5)
    assert torch.allclose(wg, w.grad, atol=1e-4)


def test_swiglu_matches_composite():
    torch.manual_seed(0)
    g = torch.randn(8, 32, requires_grad=True)
    u = torch.randn(8, 32, requires_grad=True)
    y = swiglu(g, u)
    ref = torch.nn.functional.silu(g) * u
    assert torch.allclose(y, ref, atol=1e-6)
    dy = torch.randn_like(y)
    y.backward(dy)
    gg, ug = g.grad.clone(), u.grad.clone()
    g.grad = u.grad = None
    (torch.nn.functional.silu(g) * u).backward(dy)
    assert torch.allclose(gg, g.grad, atol=1e-5)
    assert torch.allclose(ug, u.grad, atol=1e-5)


def test_rope_roundtrip_and_ref():
    torch.manual_seed(0)
    b, s, h, d = 2, 16, 4, 32
    cos, sin = build_rope_cache(s, d)
    q = torch.randn(b, s, h, d)
    k = torch.randn(b, s, 2, d)
    qo, ko = apply_rotary_pos_emb(q, k, cos, sin)
    # rotate back
    qb, kb = apply_rotary_pos_emb(qo, ko, cos, -sin)
    assert torch.allclose(qb, q, atol=1e-5)
    assert torch.allclose(kb, k, atol=1e-5)
    # norm preservation per (pair) rotation
    assert torch.allclose(qo.norm(), q.norm(), atol=1e-4)


def test_rope_grad_is_inverse_rotation():
    torch.manual_seed(0)
    cos, sin = build_rope_cache(8, 16)
    q = torch.randn(1, 8, 2, 16, requires_grad=True)
    k = torch.randn(1, 8, 2, 16, requires_grad=True)
    qo, ko = apply_rotary_pos_emb(q, k, cos, sin)
    loss = (qo.pow(2).sum() + ko.pow(2).sum())
    loss.backward()
    # d/dx |R x|^2 = 2x  (rotation preserves norm)
    assert torch.allclose(q.grad, 2 * q, atol=1e-4)


def test_cross_entropy_matches_torch():
    torch.manual_seed(0)
    logits = torch.randn(64, 100, requires_grad=True)
    target = torch.randint(0, 100, (64,))
    target[5] = -100
    loss = cross_entropy(logits, target, ignore_index=-100)
    ref = torch.nn.functional.cross_entropy(logits, target,
                                            ignore_index=-100)
    assert torch.allclose(loss, ref, atol=1e-5)
    loss.backward()
    g1 = logits.grad.clone()
    logits.grad = None
    torch.nn.functional.cross_entropy(logits, target,
                                      ignore_index=-100).backward()
    assert torch.allclose(g1, logits.grad, atol=1e-5)


def test_linear_cross_entropy_matches_full():
    torch.manual_seed(0)
    x = torch.randn(32, 64, requires_grad=True)
    w = torch.randn(200, 64, requires_grad=True)
    t = torch.randint(0, 200, (32,))
    loss = linear_cross_entropy(x, w, t)
    loss.backward()
    xg, wg = x.grad.clone(), w.grad.clone()
    x.grad = w.grad = None
    ref = torch.nn.functional.cross_entropy(x @ w.t(), t)
    assert torch.allclose(loss, ref, atol=1e-5)
    ref.backward()
    assert torch.allclose(xg, x.grad, atol=1e-5)
    assert torch.allclose(wg, w.grad, atol=1e-4)


@pytest.mark.parametrize("causal", [False, True])
@pytest.mark.parametrize("hk", [4, 1, 2])
def test_flash_attn_ref_matches_sdpa(causal, hk):
    torch.manual_seed(0)
    b, s, h, d = 2, 33, 4, 16
    q = torch.randn(b, s, h, d, requires_grad=True)
    k = torch.randn(b, s, hk, d, requires_grad=True)
    v = torch.randn(b, s, hk, d, requires_grad=True)
    out = flash_attn_xla(q, k, v, causal=causal)
    qt = q.transpose(1, 2)
    kt = k.transpose(1, 2)
    vt = v.transpose(1, 2)
    ref = torch.nn.functional.scaled_dot_product_attention(
        qt, kt, vt, is_causal=causal, enable_gqa=(hk != h))
    ref = ref.transpose(1, 2)
    assert torch.allclose(out, ref, atol=1e-4), (out - ref).abs().max()
    dy = torch.randn_like(out)
    out.backward(dy)
    g = [q.grad.clone(), k.grad.clone(), v.grad.clone()]
    q.grad = k.grad = v.grad = None
    ref2 = torch.nn.functional.scaled_dot_product_attention(
        q.transpose(1, 2), k.transpose(1, 2), v.transpose(1, 2),
        is_causal=causal, enable_gqa=(hk != h)).transpose(1, 2)
    ref2.backward(dy)
    assert torch.allclose(g[0], q.grad, atol=1e-4)
    assert torch.allclose(g[1], k.grad, atol=1e-4)
    assert torch.allclose(g[2], v.grad, atol=1e-4)


def test_flash_attn_varlen_mask():
    torch.manual_seed(0)
    b, s, h, d = 2, 24, 2, 16
    q = torch.randn(b, s, h, d)
    k = torch.randn(b, s, h, d)
    v = torch.randn(b, s, h, d)
    lens = torch.tensor([24, 17])
    mask = (torch.arange(s).unsqueeze(0) < lens.unsqueeze(1)).int()
    out = flash_attn_varlen_xla(q, k, v, attention_mask=mask, causal=True)
    # batch 1 truncated to 17 must equal standalone attention on 17 tokens
    out_b1 = flash_attn_xla(q[1:2, :17], k[1:2, :17], v[1:2, :17],
                            causal=True)
    assert torch.allclose(out[1, :17], out_b1[0], atol=1e-4)


def test_sliding_window():
    torch.manual_seed(0)
    b, s, h, d = 1, 32, 2, 16
    q = torch.randn(b, s, h, d)
    k = torch.randn(b, s, h, d)
    v = torch.randn(b, s, h, d)
    w = 8
    out = flash_attn_xla(q, k, v, causal=True, window_size=(w, 0))
    # manual masked attention
    scale = 1 / math.sqrt(d)
    qt, kt, vt = [t.permute(0, 2, 1, 3) for t in (q, k, v)]
    sc = qt @ kt.transpose(-1, -2) * scale
    i = torch.arange(s).view(s, 1)
    j = torch.arange(s).view(1, s)
    m = (j > i) | (j < i - w)
    sc = sc.masked_fill(m, float("-inf"))
    ref = (torch.softmax(sc, -1) @ vt).permute(0, 2, 1, 3)
    assert torch.allclose(out, ref, atol=1e-4)
