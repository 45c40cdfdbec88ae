"""Native Llama-family implementation on the framework's CDNA4 ops.

This is the flagship model used by bench.py (BASELINE.json: Llama-2-7B FSDP
bf16). It is written directly against torchacc_amd.ops — RMSNorm, fused RoPE,
flash attention, SwiGLU, fused-linear-cross-entropy — with the projection
GEMMs on hipBLASLt via torch.matmul. HF-transformers models are supported
separately through the patch layer (utils/patch.py); this native module
avoids HF overhead on the hot path.

Context parallelism: when a CP mode is set (Config.dist.sp), attention
dispatches to ulysses / ring / 2D FlashSequence and the inputs are expected
sequence-sharded (bench/accelerate handle the sharding).
"""
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from ..ops.cross_entropy import linear_cross_entropy
from ..ops.flash_attn import flash_attn_xla
from ..ops.linear import TunedLinear
from ..ops.rmsnorm import RMSNorm, fused_add_rms_norm
from ..ops.rope import apply_rotary_pos_emb, build_rope_cache
from ..ops.swiglu import swiglu


@dataclass
class LlamaConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    num_hidden_layers: int = 32
    num_attention_heads: int = 32
    num_key_value_heads: int = 32
    max_position_embeddings: int = 4096
    rms_norm_eps: float = 1e-5
    rope_theta: float = 10000.0
    tie_word_embeddings: bool = False
    # context-parallel attention mode: None|'ulysses'|'ring'|'2d'
    cp_mode: Optional[str] = None


def llama_2_7b(**kw) -> "LlamaConfig":
    return LlamaConfig(**kw)


def llama_2_70b(**kw) -> "LlamaConfig":
    base = dict(
        hidden_size=8192, intermediate_size=28672, num_hidden_layers=80,
        num_attention_heads=64, num_key_value_heads=8)
    base.update(kw)
    return LlamaConfig(**base)


def llama_3_8b(**kw) -> "LlamaConfig":
    """Llama-3 8B: GQA (8 kv heads), 128k vocab, rope theta 500k."""
    base = dict(
        vocab_size=128256, hidden_size=4096, intermediate_size=14336,
        num_hidden_layers=32, num_attention_heads=32, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0)
    base.update(kw)
    return LlamaConfig(**base)


def llama_3_70b(**kw) -> "LlamaConfig":
    """Llama-3 70B: GQA (8 kv heads), 128k vocab, rope theta 500k."""
    base = dict(
        vocab_size=128256, hidden_size=8192, intermediate_size=28672,
        num_hidden_layers=80, num_attention_heads=64, num_key_value_heads=8,
        max_position_embeddings=8192, rope_theta=500000.0)
    base.update(kw)
    return LlamaConfig(**base)


def llama_tiny(**kw) -> "LlamaConfig":
    """4-layer toy config (driver config #1: tiny DP=1 CPU plumbing)."""
    base = dict(
        vocab_size=1024, hidden_size=256, intermediate_size=688,
        num_hidden_layers=4, num_attention_heads=8, num_key_value_heads=8,
        max_position_embeddings=512)
    base.update(kw)
    return LlamaConfig(**base)


class LlamaAttention(nn.Module):

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.cfg = cfg
        h, hk = cfg.num_attention_heads, cfg.num_key_value_heads
        # instance attrs so tensor parallelism can shard heads per layer
        self.num_heads = h
        self.num_kv_heads = hk
        self.head_dim = cfg.hidden_size // h
        self.q_proj = TunedLinear(cfg.hidden_size, h * self.head_dim,
                                  bias=False)
        self.k_proj = TunedLinear(cfg.hidden_size, hk * self.head_dim,
                                  bias=False)
        self.v_proj = TunedLinear(cfg.hidden_size, hk * self.head_dim,
                                  bias=False)
        self.o_proj = TunedLinear(h * self.head_dim, cfg.hidden_size,
                                  bias=False)

    def forward(self, x, cos, sin):
        b, s, _ = x.shape
        h, hk = self.num_heads, self.num_kv_heads
        q = self.q_proj(x).view(b, s, h, self.head_dim)
        k = self.k_proj(x).view(b, s, hk, self.head_dim)
        v = self.v_proj(x).view(b, s, hk, self.head_dim)
        mode = self.cfg.cp_mode
        if mode is None:
            q, k = apply_rotary_pos_emb(q, k, cos, sin)
            o = flash_attn_xla(q, k, v, causal=True)
        elif mode == "ulysses":
            from ..ops.context_parallel import ulysses

            def rope_fn(qq, kk):
                return apply_rotary_pos_emb(qq, kk, cos, sin)

            o = ulysses(q, k, v, causal=True, rope_func=rope_fn)
        elif mode == "ring":
            from ..ops.context_parallel import (get_inter_cp_group,
                                                ring_attention)
            import torch.distributed as dist
            g = get_inter_cp_group()
            r = dist.get_rank(g) if g is not None else 0
            q, k = apply_rotary_pos_emb(q, k, cos[r * s:(r + 1) * s],
                                        sin[r * s:(r + 1) * s])
            o = ring_attention(q, k, v, causal=True)
        elif mode == "2d":
            from ..ops.context_parallel import (context_parallel_2d,
                                                get_context_parallel_group)
            import torch.distributed as dist
            g = get_context_parallel_group()
            r = dist.get_rank(g) if g is not None else 0
            q, k = apply_rotary_pos_emb(q, k, cos[r * s:(r + 1) * s],
                                        sin[r * s:(r + 1) * s])
            o = context_parallel_2d(q, k, v, causal=True)
        else:
            raise ValueError(f"unknown cp mode {mode}")
        return self.o_proj(o.reshape(b, s, h * self.head_dim))  # noqa


class LlamaMLP(nn.Module):

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.gate_proj = TunedLinear(cfg.hidden_size,
                                     cfg.intermediate_size, bias=False)
        self.up_proj = TunedLinear(cfg.hidden_size, cfg.intermediate_size,
                                   bias=False)
        self.down_proj = TunedLinear(cfg.intermediate_size, cfg.hidden_size,
                                     bias=False)

    def forward(self, x):
        return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


class LlamaDecoderLayer(nn.Module):
    """Residual-stream form: the layer receives (residual, delta) with
    hidden = residual + delta and returns the same pair — every residual
    add is fused into the following RMSNorm (one kernel: add + norm +
    residual-grad accumulation in backward)."""

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.self_attn = LlamaAttention(cfg)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size,
                                                cfg.rms_norm_eps)
        self.mlp = LlamaMLP(cfg)

    def forward(self, residual, delta, cos, sin):
        y1, resid = fused_add_rms_norm(
            delta, residual, self.input_layernorm.weight,
            self.input_layernorm.variance_epsilon)
        a = self.self_attn(y1, cos, sin)
        y2, resid2 = fused_add_rms_norm(
            a, resid, self.post_attention_layernorm.weight,
            self.post_attention_layernorm.variance_epsilon)
        m = self.mlp(y2)
        return resid2, m


class LlamaForCausalLM(nn.Module):

    def __init__(self, cfg: LlamaConfig):
        super().__init__()
        self.config = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [LlamaDecoderLayer(cfg) for _ in range(cfg.num_hidden_layers)])
        self.norm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.lm_head = nn.Linear(cfg.hidden_size, cfg.vocab_size, bias=False)
        if cfg.tie_word_embeddings:
            self.lm_head.weight = self.embed_tokens.weight
        head_dim = cfg.hidden_size // cfg.num_attention_heads
        cos, sin = build_rope_cache(cfg.max_position_embeddings, head_dim,
                                    cfg.rope_theta)
        self.register_buffer("rope_cos", cos, persistent=False)
        self.register_buffer("rope_sin", sin, persistent=False)

        self.apply(self._init_weights)

    def _apply(self, fn, recurse=True):
        out = super()._apply(fn, recurse)
        # .to(bfloat16)/.half() sweeps buffers too: the RoPE tables must
        # stay fp32 (FA2 semantics) — re-float after any cast/move
        self.rope_cos = self.rope_cos.float()
        self.rope_sin = self.rope_sin.float()
        return out

    @staticmethod
    def _init_weights(m):
        if isinstance(m, nn.Linear):
            nn.init.normal_(m.weight, std=0.02)
        elif isinstance(m, nn.Embedding):
            nn.init.normal_(m.weight, std=0.02)

    @torch.no_grad()
    def generate(self, input_ids: torch.Tensor, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 eos_token_id=None) -> torch.Tensor:
        """KV-cache autoregressive decode (models/generation.py)."""
        from .generation import generate as _gen
        return _gen(self, input_ids, max_new_tokens, temperature, top_k,
                    eos_token_id)

    def forward(self, input_ids: torch.Tensor,
                labels: Optional[torch.Tensor] = None,
                attention_mask=None):
        delta = self.embed_tokens(input_ids)
        residual = None
        cos, sin = self.rope_cos, self.rope_sin
        for layer in self.layers:
            residual, delta = layer(residual, delta, cos, sin)
        x, _ = fused_add_rms_norm(delta, residual, self.norm.weight,
                                  self.norm.variance_epsilon)
        if labels is not None:
            # shift: predict token t+1 from position t; fused linear+CE
            hs = x[:, :-1, :].reshape(-1, x.shape[-1])
            tg = labels[:, 1:].reshape(-1)
            if getattr(self, "_tp_group", None) is not None:
                # lm_head is vocab-sharded (dist/tp.py surgery): CE on
                # sharded logits, never materializing [N, vocab] anywhere
                from ..dist.tp import vocab_parallel_linear_cross_entropy
                return vocab_parallel_linear_cross_entropy(
                    hs, self.lm_head.weight, tg, self._tp_group)
            return linear_cross_entropy(hs, self.lm_head.weight, tg)
        return self.lm_head(x)
