"""Context parallelism parity vs single-device flash attention
(reference tests/ops/test_context_parallel.py:33-186 — whose ring/2d cases
were skipped for correctness issues; ours must pass)."""
import pytest
import torch

from tests.utils.distributed import run_multiprocess

B, S, H, D = 2, 64, 4, 32


def _full_inputs(seed=0):
    torch.manual_seed(seed)
    q = torch.randn(B, S, H, D)
    k = torch.randn(B, S, H, D)
    v = torch.randn(B, S, H, D)
    dout = torch.randn(B, S, H, D)
    return q, k, v, dout


def _single_device_ref(causal=True):
    from torchacc_amd.ops.flash_attn import flash_attn_xla
    q, k, v, dout = _full_inputs()
    q, k, v = [t.clone().requires_grad_(True) for t in (q, k, v)]
    out = flash_attn_xla(q, k, v, causal=causal)
    out.backward(dout)
    return out.detach(), q.grad, k.grad, v.grad


def _cp_worker(rank, world, mode, out_dir):
    import torchacc_amd as ta  # noqa: F401
    from torchacc_amd.ops.context_parallel import (
        context_parallel_2d, initialize_context_parallel, ring_attention,
        ulysses)
    intra = world if mode == "ulysses" else (1 if mode == "ring" else
                                             world // 2)
    initialize_context_parallel(world, intra)
    q, k, v, dout = _full_inputs()
    chunk = S // world
    sl = slice(rank * chunk, (rank + 1) * chunk)
    ql = q[:, sl].clone().requires_grad_(True)
    kl = k[:, sl].clone().requires_grad_(True)
    vl = v[:, sl].clone().requires_grad_(True)
    if mode == "ulysses":
        out = ulysses(ql, kl, vl, causal=True)
    elif mode == "ring":
        out = ring_attention(ql, kl, vl, causal=True)
    else:
        out = context_parallel_2d(ql, kl, vl, causal=True)
    out.backward(dout[:, sl])
    import numpy as np
    np.savez(f"{out_dir}/rank{rank}.npz", o=out.detach().numpy(),
             dq=ql.grad.numpy(), dk=kl.grad.numpy(), dv=vl.grad.numpy())


@pytest.mark.parametrize("mode,world", [
    ("ulysses", 2), ("ring", 2), ("ring", 4), ("2d", 4),
])
def test_cp_matches_single_device(mode, world, tmp_path):
    import numpy as np
    run_multiprocess(_cp_worker, world_size=world,
                     args=(mode, str(tmp_path)))
    ref_out, ref_dq, ref_dk, ref_dv = _single_device_ref()
    chunk = S // world
    for r in range(world):
        sl = slice(r * chunk, (r + 1) * chunk)
        z = np.load(tmp_path / f"rank{r}.npz")
        o, dq, dk, dv = [torch.from_numpy(z[n])
                         for n in ("o", "dq", "dk", "dv")]
        assert torch.allclose(o, ref_out[:, sl], atol=2e-4), \
            f"{mode} rank {r} out err " \
            f"{(o - ref_out[:, sl]).abs().max():.2e}"
        assert torch.allclose(dq, ref_dq[:, sl], atol=2e-4), \
            f"{mode} rank {r} dq"
        assert torch.allclose(dk, ref_dk[:, sl], atol=2e-4), \
            f"{mode} rank {r} dk"
        assert torch.allclose(dv, ref_dv[:, sl], atol=2e-4), \
            f"{mode} rank {r} dv"


def test_update_out_and_lse_merge():
    """Merging two disjoint key blocks reproduces full attention."""
    from torchacc_amd.ops.context_parallel.utils import update_out_and_lse
    from torchacc_amd.ops.flash_attn import _ref_attention
    torch.manual_seed(0)
    q = torch.randn(1, 8, 2, 16)
    k = torch.randn(1, 16, 2, 16)
    v = torch.randn(1, 16, 2, 16)
    scale = 16 ** -0.5
    full, full_lse = _ref_attention(q, k, v, scale, False, (-1, -1))
    o1, l1 = _ref_attention(q, k[:, :8], v[:, :8], scale, False, (-1, -1))
    o2, l2 = _ref_attention(q, k[:, 8:], v[:, 8:], scale, False, (-1, -1))
    out, lse = update_out_and_lse(None, None, o1, l1)
    out, lse = update_out_and_lse(out, lse, o2, l2)
    assert torch.allclose(out.to(full.dtype), full, atol=1e-5)
    assert torch.allclose(lse, full_lse, atol=1e-5)


LENS = [40, 24]  # per-batch true lengths (B=2, S=64 padded)


def _varlen_single_ref(causal=True):
    from torchacc_amd.ops.flash_attn import FlashAttnFunc
    q, k, v, dout = _full_inputs()
    lens = torch.tensor(LENS, dtype=torch.int32)
    pos = torch.arange(S)
    qmask = (pos.unsqueeze(0) < lens.unsqueeze(1))       # [B,S]
    dout = dout * qmask.unsqueeze(-1).unsqueeze(-1)
    q, k, v = [t.clone().requires_grad_(True) for t in (q, k, v)]
    out, _ = FlashAttnFunc.apply(q, k, v, 0.0, D ** -0.5, causal, (-1, -1),
                                 None, False, lens, lens)
    out.backward(dout)
    return out.detach(), q.grad, k.grad, v.grad, qmask


def _cp_varlen_worker(rank, world, mode, out_dir):
    import torchacc_amd as ta  # noqa: F401
    from torchacc_amd.ops.context_parallel import (
        context_parallel_2d, initialize_context_parallel, ring_attention)
    intra = 1 if mode == "ring" else world // 2
    initialize_context_parallel(world, intra)
    q, k, v, dout = _full_inputs()
    lens = torch.tensor(LENS, dtype=torch.int32)
    pos = torch.arange(S)
    qmask = (pos.unsqueeze(0) < lens.unsqueeze(1))
    dout = dout * qmask.unsqueeze(-1).unsqueeze(-1)
    chunk = S // world
    sl = slice(rank * chunk, (rank + 1) * chunk)
    ql = q[:, sl].clone().requires_grad_(True)
    kl = k[:, sl].clone().requires_grad_(True)
    vl = v[:, sl].clone().requires_grad_(True)
    if mode == "ring":
        out = ring_attention(ql, kl, vl, causal=True, q_lens=lens,
                             k_lens=lens)
    else:
        out = context_parallel_2d(ql, kl, vl, causal=True, q_lens=lens,
                                  k_lens=lens)
    out.backward(dout[:, sl])
    import numpy as np
    np.savez(f"{out_dir}/rank{rank}.npz", o=out.detach().numpy(),
             dq=ql.grad.numpy(), dk=kl.grad.numpy(), dv=vl.grad.numpy())


@pytest.mark.parametrize("mode,world", [
    ("ring", 2), ("ring", 4), ("2d", 4),
])
def test_cp_varlen_matches_single_device(mode, world, tmp_path):
    """Varlen ring / 2D: global per-batch true lengths threaded through the
    ring (reference ring_attn.py:431-508 — whose own tests were skipped).
    Rank 2+ holds blocks that are fully padding for batch 1 (zero-length
    block path)."""
    import numpy as np
    run_multiprocess(_cp_varlen_worker, world_size=world,
                     args=(mode, str(tmp_path)))
    ref_out, ref_dq, ref_dk, ref_dv, qmask = _varlen_single_ref()
    chunk = S // world
    m4 = qmask.unsqueeze(-1).unsqueeze(-1)
    for r in range(world):
        sl = slice(r * chunk, (r + 1) * chunk)
        z = np.load(tmp_path / f"rank{r}.npz")
        o, dq, dk, dv = [torch.from_numpy(z[n])
                         for n in ("o", "dq", "dk", "dv")]
        lm = m4[:, sl]
        assert torch.allclose(o * lm, ref_out[:, sl] * lm, atol=2e-4), \
            f"{mode} rank {r} out err " \
            f"{((o - ref_out[:, sl]) * lm).abs().max():.2e}"
        assert torch.allclose(dq * lm, ref_dq[:, sl] * lm, atol=2e-4), \
            f"{mode} rank {r} dq"
        assert torch.allclose(dk * lm, ref_dk[:, sl] * lm, atol=2e-4), \
            f"{mode} rank {r} dk"
        assert torch.allclose(dv * lm, ref_dv[:, sl] * lm, atol=2e-4), \
            f"{mode} rank {r} dv"
