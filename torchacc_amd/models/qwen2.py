"""Native Qwen2 family on the framework's CDNA4 ops.

Qwen2 = the Llama architecture with attention QKV biases, GQA and optional
sliding-window attention (the reference accelerated Qwen via source patches,
llm/qwen_patch.py; this is the native equivalent of models/llama.py).
"""
from dataclasses import dataclass
from typing import Optional

import torch
import torch.nn as nn

from ..ops.cross_entropy import linear_cross_entropy
from ..ops.flash_attn import flash_attn_xla
from ..ops.rmsnorm import RMSNorm, fused_add_rms_norm
from ..ops.rope import apply_rotary_pos_emb, build_rope_cache
from .llama import LlamaMLP


@dataclass
class Qwen2Config:
    vocab_size: int = 151936
    hidden_size: int = 3584
    intermediate_size: int = 18944
    num_hidden_layers: int = 28
    num_attention_heads: int = 28
    num_key_value_heads: int = 4
    max_position_embeddings: int = 4096
    rms_norm_eps: float = 1e-6
    rope_theta: float = 1000000.0
    sliding_window: Optional[int] = None
    tie_word_embeddings: bool = False


def qwen2_7b(**kw) -> Qwen2Config:
    return Qwen2Config(**kw)


def qwen2_tiny(**kw) -> Qwen2Config:
    return Qwen2Config(
        vocab_size=1024, hidden_size=256, intermediate_size=688,
        num_hidden_layers=2, num_attention_heads=8, num_key_value_heads=2,
        max_position_embeddings=512, **kw)


class Qwen2Attention(nn.Module):

    def __init__(self, cfg: Qwen2Config):
        super().__init__()
        self.cfg = cfg
        h, hk = cfg.num_attention_heads, cfg.num_key_value_heads
        self.num_heads = h
        self.num_kv_heads = hk
        self.head_dim = cfg.hidden_size // h
        self.q_proj = nn.Linear(cfg.hidden_size, h * self.head_dim, bias=True)
        self.k_proj = nn.Linear(cfg.hidden_size, hk * self.head_dim,
                                bias=True)
        self.v_proj = nn.Linear(cfg.hidden_size, hk * self.head_dim,
                                bias=True)
        from ..ops.linear import TunedLinear
        self.o_proj = TunedLinear(h * self.head_dim, cfg.hidden_size,
                                  bias=False)

    def forward(self, x, cos, sin):
        b, s, _ = x.shape
        h, hk = self.num_heads, self.num_kv_heads
        q = self.q_proj(x).view(b, s, h, self.head_dim)
        k = self.k_proj(x).view(b, s, hk, self.head_dim)
        v = self.v_proj(x).view(b, s, hk, self.head_dim)
        q, k = apply_rotary_pos_emb(q, k, cos, sin)
        window = (-1, -1)
        if self.cfg.sliding_window is not None and \
                s > self.cfg.sliding_window:
            window = (self.cfg.sliding_window, 0)
        o = flash_attn_xla(q, k, v, causal=True, window_size=window)
        return self.o_proj(o.reshape(b, s, h * self.head_dim))


class Qwen2DecoderLayer(nn.Module):

    def __init__(self, cfg: Qwen2Config):
        super().__init__()
        self.input_layernorm = RMSNorm(cfg.hidden_size, cfg.rms_norm_eps)
        self.self_attn = Qwen2Attention(cfg)
        self.post_attention_layernorm = RMSNorm(cfg.hidden_size,
                                                cfg.rms_norm_eps)
        self.mlp = LlamaMLP(cfg)  # same SwiGLU structure

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


class Qwen2ForCausalLM(nn.Module):

    def __init__(self, cfg: Qwen2Config):
        super().__init__()
        self.config = cfg
        self.embed_tokens = nn.Embedding(cfg.vocab_size, cfg.hidden_size)
        self.layers = nn.ModuleList(
            [Qwen2DecoderLayer(cfg) for _ in range(cfg.num_hidden_layers)])
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
        if isinstance(m, (nn.Linear, nn.Embedding)):
            nn.init.normal_(m.weight, std=0.02)
        if isinstance(m, nn.Linear) and m.bias is not None:
            nn.init.zeros_(m.bias)

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 32,
                 temperature: float = 0.0, top_k: int = 0,
                 eos_token_id=None):
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
            hs = x[:, :-1, :].reshape(-1, x.shape[-1])
            tg = labels[:, 1:].reshape(-1)
            if getattr(self, "_tp_group", None) is not None:
                from ..dist.tp import vocab_parallel_linear_cross_entropy
                return vocab_parallel_linear_cross_entropy(
                    hs, self.lm_head.weight, tg, self._tp_group)
            return linear_cross_entropy(hs, self.lm_head.weight, tg)
        return self.lm_head(x)
