"""Qwen remote-code patch (reference llm/qwen_patch.py:9-53).

The reference regex-rewrote the remote-code Qwen-72B attention forward to
drop CUDA/dtype asserts and swap flash_attn_unpadded_func for the
accelerated varlen kernel. Here the same goal is met by rebinding the
remote module's flash-attn symbols to this framework's ops — no source
re-exec needed on the eager backend.
"""
import sys
from ..utils.logger import logger


def patch_qwen_model(model) -> bool:
    """Point a remote-code Qwen model's flash-attn imports at our kernels."""
    from ..ops.flash_attn import (flash_attn_varlen_func, flash_attn_func)
    mod = sys.modules.get(type(model).__module__)
    if mod is None:
        return False
    patched = False
    for name, repl in (("flash_attn_unpadded_func", flash_attn_varlen_func),
                       ("flash_attn_varlen_func", flash_attn_varlen_func),
                       ("flash_attn_func", flash_attn_func)):
        if hasattr(mod, name):
            setattr(mod, name, repl)
            patched = True
    if patched:
        logger.info("patched %s flash-attn entry points",
                    type(model).__module__)
    return patched


rewrite_forward = patch_qwen_model
