"""HF-transformers integration patches.

Reimplements the intent of the reference's utils/patch.py:51-313 + ops/liger
for the eager ROCm backend:

- :func:`patch_fa`: route transformers' internal flash-attention entry point
  (``modeling_flash_attention_utils._flash_attention_forward``) to our CDNA4
  kernels, picking the variant exactly as the reference does (:100-211):
  position_ids with bsz==1 -> packed varlen; attention_mask -> varlen-by-mask;
  else fixed-length.
- :func:`apply_fused_kernel_patches`: swap HF Llama/Qwen2 RMSNorm / SwiGLU-MLP
  forwards for our fused kernels (the reference's Liger integration,
  ops/liger.py:10-130).
- :func:`patch_llama` / :func:`patch_qwen`: disable mask materialization
  (the `_update_causal_mask -> None` trick, reference :224-301) since flash
  attention builds no mask.
- :func:`patch_amp`: map torch.optim.{Adam,AdamW,SGD} to our fused syncfree
  optimizers (reference :51-58).

All patches are gated on transformers being importable and are silent no-ops
otherwise. ``TORCHACC_PATCH_FA=0`` disables patch_fa (reference :66).
"""
import os
from typing import Optional

import torch

from .logger import logger


def _transformers():
    try:
        import transformers
        return transformers
    except ImportError:
        return None


def patch_fa() -> bool:
    tf = _transformers()
    if tf is None or os.environ.get("TORCHACC_PATCH_FA", "1") == "0":
        return False
    try:
        from transformers import modeling_flash_attention_utils as mfa
    except ImportError:
        return False

    from ..ops.flash_attn import (flash_attn_varlen_position_ids_xla,
                                  flash_attn_varlen_xla, flash_attn_xla)

    def _flash_attention_forward(query_states, key_states, value_states,
                                 attention_mask, query_length,
                                 is_causal=True, dropout=0.0,
                                 position_ids=None, softmax_scale=None,
                                 sliding_window=None,
                                 use_top_left_mask=False,
                                 softcap=None, deterministic=None,
                                 **kwargs):
        causal = is_causal
        window = (-1, -1)
        if sliding_window is not None and \
                key_states.shape[1] > sliding_window:
            window = (sliding_window, sliding_window)
        if position_ids is not None and query_states.shape[0] == 1 and \
                (torch.diff(position_ids.reshape(-1)) < 0).any():
            out = flash_attn_varlen_position_ids_xla(
                query_states, key_states, value_states, position_ids,
                dropout_p=dropout, softmax_scale=softmax_scale,
                causal=causal, window_size=window)
        elif attention_mask is not None:
            out = flash_attn_varlen_xla(
                query_states, key_states, value_states,
                attention_mask=attention_mask, dropout_p=dropout,
                softmax_scale=softmax_scale, causal=causal,
                window_size=window)
        else:
            out = flash_attn_xla(
                query_states, key_states, value_states, dropout_p=dropout,
                softmax_scale=softmax_scale, causal=causal,
                window_size=window)
        return out

    mfa._flash_attention_forward = _flash_attention_forward
    # the integrations module binds the symbol by value at import time —
    # rebind it there too so real model forwards hit our kernels
    try:
        from transformers.integrations import flash_attention as _fa_int
        _fa_int._flash_attention_forward = _flash_attention_forward
    except ImportError:
        pass
    # let users request attn_implementation="flash_attention_2" without the
    # CUDA flash_attn package: our CDNA4 kernels ARE the implementation
    matrix = getattr(mfa, "FLASH_ATTENTION_COMPATIBILITY_MATRIX", None)
    if isinstance(matrix, dict) and 2 in matrix:
        matrix[2]["general_availability_check"] = lambda *a, **k: True
        matrix[2]["pkg_availability_check"] = lambda *a, **k: True
    # transformers >= 5 lazily resolves flash_attn_func/flash_attn_varlen_fn
    # globals at model init (lazy_import_flash_attention) — pre-populate
    # them with adapters over our kernels so construction with
    # attn_implementation="flash_attention_2" needs no flash_attn package
    if hasattr(mfa, "_loaded_implementation"):
        from ..ops.flash_attn import flash_attn_varlen_func as _ta_varlen

        def _ta_fa_func(q, k, v, dropout_p=0.0, softmax_scale=None,
                        causal=False, window_size=(-1, -1),
                        alibi_slopes=None, deterministic=False,
                        return_attn_probs=False, **kw):
            return flash_attn_xla(q, k, v, dropout_p=dropout_p,
                                  softmax_scale=softmax_scale, causal=causal,
                                  window_size=window_size,
                                  alibi_slopes=alibi_slopes)

        def _ta_fa_varlen_func(q, k, v, cu_seqlens_q, cu_seqlens_k,
                               max_seqlen_q, max_seqlen_k, dropout_p=0.0,
                               softmax_scale=None, causal=False,
                               window_size=(-1, -1), alibi_slopes=None,
                               deterministic=False, return_attn_probs=False,
                               **kw):
            return _ta_varlen(q, k, v, cu_seqlens_q, cu_seqlens_k,
                              max_seqlen_q, max_seqlen_k,
                              dropout_p=dropout_p,
                              softmax_scale=softmax_scale, causal=causal,
                              window_size=window_size)

        try:
            mfa._loaded_implementation = "flash_attention_2"
            mfa._flash_fn = _ta_fa_func
            mfa._flash_varlen_fn = _ta_fa_varlen_func
            if hasattr(mfa, "_lazy_define_process_function"):
                mfa._process_flash_kwargs_fn = \
                    mfa._lazy_define_process_function(_ta_fa_varlen_func)
        except Exception:  # version drift: our _flash_attention_forward
            pass           # replacement still covers the runtime path
    logger.info("patched transformers flash-attention entry point")
    return True


def _patched_rmsnorm_forward(self, hidden_states):
    from ..ops.rmsnorm import rms_norm
    return rms_norm(hidden_states, self.weight, self.variance_epsilon)


def _is_silu(act) -> bool:
    if isinstance(act, torch.nn.SiLU):
        return True
    name = getattr(act, "__name__", type(act).__name__).lower()
    return name in ("silu", "swish", "silu_", "silukernel")


def _patched_mlp_forward(self, x):
    act = getattr(self, "act_fn", None)
    if act is not None and not _is_silu(act):
        # honor config.hidden_act: only the SiLU gate is fused
        return self.down_proj(act(self.gate_proj(x)) * self.up_proj(x))
    from ..ops.swiglu import swiglu
    return self.down_proj(swiglu(self.gate_proj(x), self.up_proj(x)))


_ORIG_CAUSAL_FWD = {}


def _lce_causal_forward(self, input_ids=None, attention_mask=None,
                        position_ids=None, past_key_values=None,
                        inputs_embeds=None, labels=None, use_cache=None,
                        output_attentions=None, output_hidden_states=None,
                        return_dict=None, cache_position=None,
                        logits_to_keep=0, **kwargs):
    """HF causal-LM forward with the loss fused into the lm_head GEMM —
    the full [b, s, vocab] logits tensor is never materialized (Liger
    lce_forward parity, reference ops/liger.py:75-76). ``logits`` in the
    returned output is None when labels are given, exactly like Liger."""
    orig = _ORIG_CAUSAL_FWD[type(self)]
    num_items = kwargs.pop("num_items_in_batch", None)
    if labels is None or output_attentions or output_hidden_states or \
            logits_to_keep not in (0, None):
        return orig(self, input_ids=input_ids,
                    attention_mask=attention_mask,
                    position_ids=position_ids,
                    past_key_values=past_key_values,
                    inputs_embeds=inputs_embeds, labels=labels,
                    use_cache=use_cache,
                    output_attentions=output_attentions,
                    output_hidden_states=output_hidden_states,
                    return_dict=return_dict, cache_position=cache_position,
                    logits_to_keep=logits_to_keep or 0, **kwargs)
    outputs = self.model(
        input_ids=input_ids, attention_mask=attention_mask,
        position_ids=position_ids, past_key_values=past_key_values,
        inputs_embeds=inputs_embeds, use_cache=use_cache,
        cache_position=cache_position, **kwargs)
    hidden = getattr(outputs, "last_hidden_state", None)
    if hidden is None:
        hidden = outputs[0]
    hs = hidden[:, :-1, :].reshape(-1, hidden.shape[-1])
    tg = labels[:, 1:].reshape(-1)
    from ..ops.cross_entropy import linear_cross_entropy
    loss = linear_cross_entropy(hs, self.lm_head.weight, tg)
    if num_items is not None:
        # HF Trainer gradient-accumulation normalization: divide the token
        # SUM by the global token count instead of the local mean
        nvalid = (tg != -100).sum().clamp_min(1)
        loss = loss * nvalid.to(loss.dtype) / num_items
    from transformers.modeling_outputs import CausalLMOutputWithPast
    return CausalLMOutputWithPast(
        loss=loss, logits=None,
        past_key_values=getattr(outputs, "past_key_values", None),
        hidden_states=getattr(outputs, "hidden_states", None),
        attentions=None)


def _hf_rope_forward(q, k, cos, sin, position_ids=None, unsqueeze_dim=1):
    """HF apply_rotary_pos_emb signature (q/k [b, h, s, d]; cos/sin
    [b, s, d] duplicated halves) over the fused CDNA4 RoPE kernel."""
    import torch.nn.functional  # noqa: F401 (parity with HF imports)
    if cos.dim() != 3 or cos.shape[0] != 1 or unsqueeze_dim != 1:
        # per-batch position tables (packing): keep HF's eager math
        from transformers.models.llama.modeling_llama import rotate_half
        c = cos.unsqueeze(unsqueeze_dim)
        s = sin.unsqueeze(unsqueeze_dim)
        return q * c + rotate_half(q) * s, k * c + rotate_half(k) * s
    d2 = q.shape[-1] // 2
    from ..ops.rope import _RoPE
    qo, ko = _RoPE.apply(q.permute(0, 2, 1, 3), k.permute(0, 2, 1, 3),
                         cos[0, :, :d2], sin[0, :, :d2])
    return qo.permute(0, 2, 1, 3), ko.permute(0, 2, 1, 3)


def apply_fused_kernel_patches(model: Optional[torch.nn.Module] = None
                               ) -> bool:
    """Swap HF Llama/Qwen2 RMSNorm / SwiGLU-MLP / RoPE forwards for the
    fused CDNA4 kernels and fuse lm_head+loss into chunked
    linear-cross-entropy (Liger-equivalent, reference ops/liger.py:133)."""
    tf = _transformers()
    if tf is None:
        return False
    patched = False
    for mod_name, rms_name, mlp_name, lm_name in (
            ("transformers.models.llama.modeling_llama", "LlamaRMSNorm",
             "LlamaMLP", "LlamaForCausalLM"),
            ("transformers.models.qwen2.modeling_qwen2", "Qwen2RMSNorm",
             "Qwen2MLP", "Qwen2ForCausalLM")):
        try:
            import importlib
            m = importlib.import_module(mod_name)
        except ImportError:
            continue
        rms = getattr(m, rms_name, None)
        if rms is not None:
            rms.forward = _patched_rmsnorm_forward
            patched = True
        mlp = getattr(m, mlp_name, None)
        if mlp is not None:
            mlp.forward = _patched_mlp_forward
            patched = True
        lm = getattr(m, lm_name, None)
        if lm is not None and lm not in _ORIG_CAUSAL_FWD:
            _ORIG_CAUSAL_FWD[lm] = lm.forward
            lm.forward = _lce_causal_forward
        if hasattr(m, "apply_rotary_pos_emb"):
            m.apply_rotary_pos_emb = _hf_rope_forward
    if patched:
        logger.info("applied fused RMSNorm/SwiGLU/RoPE/linear-CE patches "
                    "to HF models")
    return patched


# reference-compatible aliases (ops/liger.py:133 exposed per-model names)
apply_liger_kernel = apply_fused_kernel_patches
apply_liger_kernel_to_llama = apply_fused_kernel_patches
apply_liger_kernel_to_qwen2 = apply_fused_kernel_patches


def patch_llama(use_flash_attn: bool = True) -> bool:
    tf = _transformers()
    if tf is None:
        return False
    try:
        from transformers.models.llama import modeling_llama
    except ImportError:
        return False
    if hasattr(modeling_llama, "LlamaModel"):
        modeling_llama.LlamaModel._update_causal_mask = \
            lambda self, *a, **k: None
    if use_flash_attn:
        patch_fa()
    return True


def patch_qwen(use_flash_attn: bool = True) -> bool:
    tf = _transformers()
    if tf is None:
        return False
    try:
        from transformers.models.qwen2 import modeling_qwen2
    except ImportError:
        return False
    if hasattr(modeling_qwen2, "Qwen2Model"):
        modeling_qwen2.Qwen2Model._update_causal_mask = \
            lambda self, *a, **k: None
    if use_flash_attn:
        patch_fa()
    return True


def patch_amp() -> None:
    """torch.optim.{AdamW,Adam} -> fused syncfree AdamW
    (reference patch.py:51-58)."""
    from ..ops.adamw import AdamW
    torch.optim.AdamW = AdamW
    logger.info("patched torch.optim.AdamW -> torchacc_amd.ops.AdamW")


def patch_autocast() -> None:
    """No-op on the eager backend (the reference needed to redirect
    torch.autocast('xla'); there is no xla device here)."""
