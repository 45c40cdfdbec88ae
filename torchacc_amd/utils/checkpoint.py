"""Gradient (activation) checkpointing (reference utils/checkpoint.py:17-81).

Wraps modules whose class name is in ``gc_cls`` (or the root model when the
set is empty) with torch.utils.checkpoint (non-reentrant: plays well with the
FSDP engine's pre-backward re-gather hooks). ``gc_cnt`` limits how many
instances are wrapped (reference fsdp.py:182-194 interleave semantics).
"""
from typing import Optional, Set

import torch
from torch.utils.checkpoint import checkpoint

from .logger import logger


_CKPT_PREFIX = "_checkpoint_wrapped_module."


def strip_checkpoint_prefix(name: str) -> str:
    """Remove every '_checkpoint_wrapped_module.' segment from a dotted
    parameter/buffer name so state dicts are identical with and without
    gradient checkpointing (and match the reference checkpoint format)."""
    return name.replace(_CKPT_PREFIX, "")


class CheckpointWrapper(torch.nn.Module):

    def __init__(self, mod: torch.nn.Module, selective_attn: bool = False):
        super().__init__()
        self._checkpoint_wrapped_module = mod
        self._selective_attn = selective_attn
        self._register_state_dict_hook(self._strip_prefix_hook)
        self._register_load_state_dict_pre_hook(
            self._add_prefix_hook, with_module=True)

    @staticmethod
    def _strip_prefix_hook(module, state_dict, prefix, local_metadata):
        wrapped = prefix + _CKPT_PREFIX
        for key in list(state_dict.keys()):
            if key.startswith(wrapped):
                state_dict[prefix + key[len(wrapped):]] = state_dict.pop(key)
        return state_dict

    @staticmethod
    def _add_prefix_hook(module, state_dict, prefix, *args):
        wrapped = prefix + _CKPT_PREFIX
        for key in list(state_dict.keys()):
            if key.startswith(prefix) and not key.startswith(wrapped):
                state_dict[wrapped + key[len(prefix):]] = state_dict.pop(key)

    def forward(self, *args, **kwargs):
        if torch.is_grad_enabled():
            if self._selective_attn:
                # Megatron-style selective AC: attention outputs (out, lse)
                # are retained during the first pass and replayed during
                # recomputation (ops/flash_attn.py sac_contexts) — the
                # recompute skips the attention kernels; 288 GB HBM3E makes
                # the retained outputs cheap relative to the saved compute
                from ..ops.flash_attn import sac_contexts
                return checkpoint(self._checkpoint_wrapped_module, *args,
                                  use_reentrant=False,
                                  context_fn=sac_contexts, **kwargs)
            return checkpoint(self._checkpoint_wrapped_module, *args,
                              use_reentrant=False, **kwargs)
        return self._checkpoint_wrapped_module(*args, **kwargs)

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self._checkpoint_wrapped_module, name)


def gradient_checkpoint(model: torch.nn.Module,
                        gc_cls: Optional[Set[str]] = None,
                        gc_cnt: Optional[int] = None,
                        selective_attn: bool = False) -> torch.nn.Module:
    gc_cls = set(gc_cls or ())
    if not gc_cls:
        return CheckpointWrapper(model, selective_attn)
    targets = []
    for mod in model.modules():
        if isinstance(mod, CheckpointWrapper):
            continue
        for name, child in mod.named_children():
            if child.__class__.__name__ in gc_cls and \
                    not isinstance(child, CheckpointWrapper):
                targets.append((mod, name, child))
    count = 0
    for mod, name, child in targets:
        if gc_cnt is not None and count >= gc_cnt:
            break
        setattr(mod, name, CheckpointWrapper(child, selective_attn))
        count += 1
    logger.info("gradient checkpointing: wrapped %d modules (%s)", count,
                sorted(gc_cls))
    return model
