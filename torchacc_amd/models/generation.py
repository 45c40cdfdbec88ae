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


class GraphDecoder:
    """hipGraph-captured steady-state decode.

    The eager decode loop is launch-bound (~350 small launches per token on
    Llama-2-7B). This captures ONE whole decode step — embed, all layers
    with in-place KV writes, final norm, lm_head, argmax, and the in-graph
    position/length increments — into a hipGraph over STATIC shapes:
    attention runs against the full preallocated cache with a device-side
    ``k_lens`` bound (the varlen kernel path), RoPE tables are gathered by
    a device position index, and the next token feeds back into the input
    buffer inside the graph. Replaying the graph N times decodes N tokens
    with no host work in the loop (greedy only; sampling/eos use the eager
    path).
    """

    def __init__(self, model, b: int, max_len: int):
        p = next(model.parameters())
        assert p.is_cuda, "GraphDecoder requires a GPU model"
        self.model = model
        self.max_len = max_len
        d = model.layers[0].self_attn.head_dim
        self.caches = [
            LayerKV(b, max_len, layer.self_attn.num_kv_heads, d, p.dtype,
                    p.device)
            for layer in model.layers
        ]
        dev = p.device
        self.ids = torch.zeros(b, 1, dtype=torch.long, device=dev)
        self.pos = torch.zeros(1, dtype=torch.long, device=dev)
        self.k_lens = torch.zeros(b, dtype=torch.int32, device=dev)
        self.graph = None

    def _step_static(self):
        m = self.model
        cos = m.rope_cos.index_select(0, self.pos)
        sin = m.rope_sin.index_select(0, self.pos)
        delta = m.embed_tokens(self.ids)
        residual = None
        b, s = self.ids.shape
        for layer, cache in zip(m.layers, self.caches):
            y1, resid = fused_add_rms_norm(
                delta, residual, layer.input_layernorm.weight,
                layer.input_layernorm.variance_epsilon)
            attn = layer.self_attn
            h, hk = attn.num_heads, attn.num_kv_heads
            q = attn.q_proj(y1).view(b, s, h, attn.head_dim)
            k = attn.k_proj(y1).view(b, s, hk, attn.head_dim)
            v = attn.v_proj(y1).view(b, s, hk, attn.head_dim)
            q, k = apply_rotary_pos_emb(q, k, cos, sin)
            cache.k.index_copy_(1, self.pos, k)
            cache.v.index_copy_(1, self.pos, v)
            from ..ops.flash_attn import FlashAttnFunc
            o, _ = FlashAttnFunc.apply(
                q, cache.k, cache.v, 0.0, attn.head_dim ** -0.5, True,
                (-1, -1), None, False, None, self.k_lens)
            a = attn.o_proj(o.reshape(b, s, h * attn.head_dim))
            y2, resid2 = fused_add_rms_norm(
                a, resid, layer.post_attention_layernorm.weight,
                layer.post_attention_layernorm.variance_epsilon)
            delta = layer.mlp(y2)
            residual = resid2
        x, _ = fused_add_rms_norm(delta, residual, m.norm.weight,
                                  m.norm.variance_epsilon)
        logits = m.lm_head(x[:, -1])
        nxt = logits.argmax(-1, keepdim=True)
        # in-graph feedback: next replay consumes this token at pos+1
        self.ids.copy_(nxt)
        self.pos.add_(1)
        self.k_lens.add_(1)
        return nxt

    @torch.no_grad()
    def decode(self, prompt_ids: torch.Tensor, max_new_tokens: int):
        b, s0 = prompt_ids.shape
        assert s0 + max_new_tokens <= self.max_len
        for c in self.caches:
            c.len = 0
        # prefill eagerly (dynamic shapes), fill the static caches
        logits = _forward_step(self.model, prompt_ids, self.caches, 0)
        for c in self.caches:
            c.len = self.max_len  # cache buffers are written in place now
        first = logits.argmax(-1, keepdim=True)
        tokens = [first]
        if max_new_tokens == 1:
            return torch.cat([prompt_ids] + tokens, dim=1)
        self.ids.copy_(first)
        self.pos.fill_(s0)
        self.k_lens.fill_(s0 + 1)      # the step's own token is visible
        if self.graph is None:
            # warm twice on a side stream, then capture one step (capture
            # RECORDS the kernels; nothing executes and no state mutates
            # until the first replay)
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(2):
                    self._step_static()
            torch.cuda.current_stream().wait_stream(side)
            # rewind state mutated by the warmup
            self.ids.copy_(first)
            self.pos.fill_(s0)
            self.k_lens.fill_(s0 + 1)
            self.graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(self.graph):
                self._out = self._step_static()
        for _ in range(max_new_tokens - 1):
            self.graph.replay()
            tokens.append(self._out.clone())
        return torch.cat([prompt_ids] + tokens, dim=1)
