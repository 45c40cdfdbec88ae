"""Drop-in F.scaled_dot_product_attention replacement routed to our flash
attention (reference ops/scaled_dot_product_attention.py:7-20, installed by
accelerate() when ``compute.acc_scaled_dot_attn``).

SDPA layout is [b, h, s, d]; the flash op takes [b, s, h, d].
"""
import torch

from .flash_attn import flash_attn_xla


def scaled_dot_product_attention(query, key, value, attn_mask=None,
                                 dropout_p=0.0, is_causal=False, scale=None,
                                 enable_gqa=False):
    if attn_mask is not None or query.dtype not in (torch.float16,
                                                    torch.bfloat16):
        # semantics we do not accelerate -> genuine torch implementation
        return torch.nn.functional.scaled_dot_product_attention(
            query, key, value, attn_mask=attn_mask, dropout_p=dropout_p,
            is_causal=is_causal, scale=scale)
    q = query.transpose(1, 2)
    k = key.transpose(1, 2)
    v = value.transpose(1, 2)
    out = flash_attn_xla(q, k, v, dropout_p=dropout_p, softmax_scale=scale,
                         causal=is_causal)
    return out.transpose(1, 2)
