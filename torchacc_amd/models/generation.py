"""KV-cache autoregressive generation for the native model families.

The reference is a training-acceleration framework (no generate path);
this adds the serving-side decode loop natively on the same CDNA4 kernels:
prefill runs the fused varlen-capable flash attention over the prompt,
decode steps run single-query attention against the preallocated KV cache
(bottom-right-aligned causal with sq=1 attends the whole cache), RoPE is
applied at absolute positions via table slices, and the logits come from
one lm_head GEMM over the last position only.

Decode currently materializes a contiguous copy of the live cache slice
per step (the attention kernels take contiguous [b, s, h, d]); a
batch-strided kernel path / paged KV layout is the next serving
optimization.

Usage::

    model = LlamaForCausalLM(llama_2_7b()).cuda().to(torch.bfloat16)
    out = model.generate(input_ids, max_new_tokens=64)   # [b, s+64]
"""
from typing import List, Optional

import torch

from ..ops.flash_attn import flash_attn_xla
from ..ops.rmsnorm import fused_add_rms_norm
from ..ops.rope import apply_rotary_pos_emb


class LayerKV:
    """Preallocated per-layer cache: k/v [b, max_len, h_kv, d]."""

    def __init__(self, b: int, max_len: int, hk: int, d: int, dtype, device):
        self.k = torch.empty(b, max_len, hk, d, dtype=dtype, device=device)
        self.v = torch.empty(b, max_len, hk, d, dtype=dtype, device=device)
        self.len = 0

    def append(self, k: torch.Tensor, v: torch.Tensor):
        s = k.shape[1]
        self.k[:, self.len:self.len + s] = k
        self.v[:, self.len:self.len + s] = v
        self.len += s

    def view(self):
        return self.k[:, :self.len], self.v[:, :self.len]


def _attn_with_cache(attn, x, cos, sin, cache: LayerKV, window=(-1, -1)):
    b, s, _ = x.shape
    h, hk = attn.num_heads, attn.num_kv_heads
    q = attn.q_proj(x).view(b, s, h, attn.head_dim)
    k = attn.k_proj(x).view(b, s, hk, attn.head_dim)
    v = attn.v_proj(x).view(b, s, hk, attn.head_dim)
    q, k = apply_rotary_pos_emb(q, k, cos, sin)
    cache.append(k, v)
    kc, vc = cache.view()
    # bottom-right causal: the s new queries attend all cached keys up to
    # their own position
    o = flash_attn_xla(q, kc, vc, causal=True, window_size=window)
    return attn.o_proj(o.reshape(b, s, h * attn.head_dim))


def _forward_step(model, ids: torch.Tensor, caches: List[LayerKV],
                  pos: int) -> torch.Tensor:
    """Run [b, s] new tokens at absolute positions [pos, pos+s) through the
    model with caches; returns last-position logits [b, vocab]."""
    s = ids.shape[1]
    cos = model.rope_cos[pos:pos + s]
    sin = model.rope_sin[pos:pos + s]
    delta = model.embed_tokens(ids)
    residual = None
    for layer, cache in zip(model.layers, caches):
        y1, resid = fused_add_rms_norm(
            delta, residual, layer.input_layernorm.weight,
            layer.input_layernorm.variance_epsilon)
        window = (-1, -1)
        sw = getattr(getattr(layer.self_attn, "cfg", None),
                     "sliding_window", None)
        if sw is not None and cache.len + s > sw:
            window = (sw, 0)
        a = _attn_with_cache(layer.self_attn, y1, cos, sin, cache, window)
        y2, resid2 = fused_add_rms_norm(
            a, resid, layer.post_attention_layernorm.weight,
            layer.post_attention_layernorm.variance_epsilon)
        delta = layer.mlp(y2)
        residual = resid2
    x, _ = fused_add_rms_norm(delta, residual, model.norm.weight,
                              model.norm.variance_epsilon)
    return model.lm_head(x[:, -1:]).squeeze(1)


@torch.no_grad()
def generate(model, input_ids: torch.Tensor, max_new_tokens: int = 32,
             temperature: float = 0.0, top_k: int = 0,
             eos_token_id: Optional[int] = None) -> torch.Tensor:
    """Greedy (temperature=0) or sampled decode; returns [b, s + new]."""
    assert input_ids.dim() == 2
    b, s0 = input_ids.shape
    cfg = model.config
    max_len = s0 + max_new_tokens
    assert max_len <= cfg.max_position_embeddings, \
        (f"{max_len} tokens exceed max_position_embeddings="
         f"{cfg.max_position_embeddings}")
    p = next(model.parameters())
    caches = [
        LayerKV(b, max_len, layer.self_attn.num_kv_heads,
                layer.self_attn.head_dim, p.dtype, p.device)
        for layer in model.layers
    ]
    out = input_ids
    logits = _forward_step(model, input_ids, caches, 0)
    finished = torch.zeros(b, dtype=torch.bool, device=input_ids.device)
    for i in range(max_new_tokens):
        if temperature > 0.0:
            lg = logits.float() / temperature
            if top_k > 0:
                kth = lg.topk(top_k, dim=-1).values[:, -1:]
                lg = lg.masked_fill(lg < kth, float("-inf"))
            probs = lg.softmax(-1)
            nxt = torch.multinomial(probs, 1)
        else:
            nxt = logits.argmax(-1, keepdim=True)
        if eos_token_id is not None:
            nxt = torch.where(finished.unsqueeze(1),
                              torch.full_like(nxt, eos_token_id), nxt)
            finished |= nxt.squeeze(1) == eos_token_id
        out = torch.cat([out, nxt], dim=1)
        if eos_token_id is not None and bool(finished.all()):
            break
        if i + 1 < max_new_tokens:
            logits = _forward_step(model, nxt, caches, out.shape[1] - 1)
    return out
