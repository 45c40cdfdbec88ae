"""DeepSpeed-Ulysses sequence parallelism (reference ulysses.py:9-77).

q/k/v arrive sequence-sharded [b, s/cp, h, d]; an all-to-all over the intra
group scatters heads / gathers sequence -> [b, s, h/cp, d]; full-sequence
flash attention runs locally; a mirrored all-to-all restores the layout.
Over 8xMI355X the two a2a's ride the full 7-link xGMI fan-out.
"""
from typing import Callable, Optional

import torch

from ..flash_attn import flash_attn_xla
from .init_group import get_intra_cp_group
from .utils import diff_all_to_all


def ulysses(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
            softmax_scale=None, causal: bool = True, dropout_p: float = 0.0,
            window_size=(-1, -1),
            rope_func: Optional[Callable] = None,
            process_group=None, attention_mask=None,
            **rope_kwargs) -> torch.Tensor:
    """``attention_mask`` [b, s_full] (int, right padding over the FULL
    sequence) enables varlen attention after the sequence gather."""
    group = process_group if process_group is not None \
        else get_intra_cp_group()
    if group is None:
        if rope_func is not None:
            q, k = rope_func(q, k, **rope_kwargs)
        if attention_mask is not None:
            from ..flash_attn import flash_attn_varlen_xla
            return flash_attn_varlen_xla(
                q, k, v, attention_mask=attention_mask, dropout_p=dropout_p,
                softmax_scale=softmax_scale, causal=causal,
                window_size=window_size)
        return flash_attn_xla(q, k, v, dropout_p=dropout_p,
                              softmax_scale=softmax_scale, causal=causal,
                              window_size=window_size)
    import torch.distributed as dist
    cp = dist.get_world_size(group)
    assert q.shape[2] % cp == 0, \
        f"num q heads {q.shape[2]} not divisible by ulysses degree {cp}"
    assert k.shape[2] % cp == 0, \
        f"num kv heads {k.shape[2]} not divisible by ulysses degree {cp}"
    # [b, s/cp, h, d] -> [b, s, h/cp, d]
    q = diff_all_to_all(q, 2, 1, group)
    k = diff_all_to_all(k, 2, 1, group)
    v = diff_all_to_all(v, 2, 1, group)
    if rope_func is not None:
        # RoPE applied post-a2a so positions cover the full sequence
        q, k = rope_func(q, k, **rope_kwargs)
    if attention_mask is not None:
        from ..flash_attn import flash_attn_varlen_xla
        out = flash_attn_varlen_xla(
            q, k, v, attention_mask=attention_mask, dropout_p=dropout_p,
            softmax_scale=softmax_scale, causal=causal,
            window_size=window_size)
    else:
        out = flash_attn_xla(q, k, v, dropout_p=dropout_p,
                             softmax_scale=softmax_scale, causal=causal,
                             window_size=window_size)
    # [b, s, h/cp, d] -> [b, s/cp, h, d]
    return diff_all_to_all(out, 1, 2, group)
