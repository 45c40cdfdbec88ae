"""Ops facade (reference torchacc/ops/__init__.py:1-6)."""
from .flash_attn import (  # noqa: F401
    flash_attn_func, flash_attn_kvpacked_xla, flash_attn_qkvpacked_xla,
    flash_attn_varlen_func, flash_attn_varlen_position_ids_xla,
    flash_attn_varlen_qkvpacked_xla, flash_attn_varlen_xla, flash_attn_xla,
    spmd_flash_attn_varlen_xla)
from .rmsnorm import RMSNorm, rms_norm  # noqa: F401
from .rope import apply_rotary_pos_emb, build_rope_cache  # noqa: F401
from .swiglu import swiglu  # noqa: F401
from .cross_entropy import cross_entropy, linear_cross_entropy  # noqa: F401
from .adamw import AdamW  # noqa: F401
from .scaled_dot_product_attention import scaled_dot_product_attention  # noqa: F401

from . import context_parallel  # noqa: F401
from .context_parallel import (  # noqa: F401
    context_parallel_2d, ring_attention, ulysses)

# reference-compatible fused-kernel entry points (reference ops/liger.py;
# ours are HIP kernels, not Triton)
from ..utils.patch import (apply_liger_kernel,  # noqa: F401,E402
                           apply_liger_kernel_to_llama,
                           apply_liger_kernel_to_qwen2,
                           apply_fused_kernel_patches)
