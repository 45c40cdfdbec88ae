"""2D "FlashSequence" context parallelism (reference context_parallel_2d.py).

Hybrid: intra-node Ulysses all-to-all (head dim, bandwidth-heavy over the
7-link xGMI fan-out) wrapped around inter-node ring attention (sequence dim,
latency-tolerant overlapped P2P). Degenerates to pure Ulysses or pure ring
when the other group has size 1 (reference :99-126).

Sequence layout: the full sequence is split over all cp = inter*intra ranks;
rank (i, j) in the [inter, intra] grid holds chunk index i*intra + j. The
intra a2a gathers the intra-group's chunks (contiguous inside one inter
shard) and scatters heads; ring attention then runs over the inter group on
the head-sharded, intra-gathered sequence.
"""
import torch
import torch.distributed as dist

from .init_group import get_inter_cp_group, get_intra_cp_group
from .ring_attn import ring_attention
from .ulysses import ulysses
from .utils import diff_all_to_all


def context_parallel_2d(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                        softmax_scale=None, causal: bool = True,
                        intra_group=None, inter_group=None,
                        q_lens=None, k_lens=None) -> torch.Tensor:
    """q/k/v [b, s/cp, h, d] -> out [b, s/cp, h, d]. ``q_lens``/``k_lens``
    are GLOBAL per-batch true lengths (varlen packing over the full
    sequence); the head a2a leaves sequence ownership unchanged, so they
    pass straight through to the ring (reference ring_attn.py:431-508
    true_k_lens threading)."""
    intra_group = intra_group if intra_group is not None \
        else get_intra_cp_group()
    inter_group = inter_group if inter_group is not None \
        else get_inter_cp_group()
    intra = dist.get_world_size(intra_group) if intra_group is not None else 1
    inter = dist.get_world_size(inter_group) if inter_group is not None else 1
    def _mask_from_lens(lens, s_full):
        if lens is None:
            return None
        import torch as _t
        pos = _t.arange(s_full, device=q.device)
        return (pos.unsqueeze(0) < lens.unsqueeze(1)).to(_t.int32)

    if intra == 1 and inter == 1:
        from ..flash_attn import flash_attn_varlen_xla, flash_attn_xla
        if k_lens is not None:
            return flash_attn_varlen_xla(
                q, k, v, attention_mask=_mask_from_lens(k_lens, k.shape[1]),
                softmax_scale=softmax_scale, causal=causal)
        return flash_attn_xla(q, k, v, softmax_scale=softmax_scale,
                              causal=causal)
    if inter == 1:
        return ulysses(q, k, v, softmax_scale=softmax_scale, causal=causal,
                       process_group=intra_group,
                       attention_mask=_mask_from_lens(
                           k_lens, k.shape[1] * intra))
    if intra == 1:
        return ring_attention(q, k, v, softmax_scale=softmax_scale,
                              causal=causal, process_group=inter_group,
                              q_lens=q_lens, k_lens=k_lens)
    # intra a2a: [b, s/cp, h, d] -> [b, s/inter, h/intra, d]
    q = diff_all_to_all(q, 2, 1, intra_group)
    k = diff_all_to_all(k, 2, 1, intra_group)
    v = diff_all_to_all(v, 2, 1, intra_group)
    out = ring_attention(q, k, v, softmax_scale=softmax_scale, causal=causal,
                         process_group=inter_group, q_lens=q_lens,
                         k_lens=k_lens)
    return diff_all_to_all(out, 1, 2, intra_group)
