"""fx tracing for pipeline-parallel graph splitting.

Reimplements the intent of the reference's utils/trace.py:21-176: trace with
``concrete_args`` derived from the forward signature (unused-by-name inputs
become concrete defaults), using HF's HFTracer for PreTrainedModel and a
plain ``torch.fx.Tracer`` otherwise. The framework's fused ops are traced as
leaf call_function nodes (registered below) so autograd.Function internals
never enter the graph.
"""
import inspect
from typing import List, Optional

import torch
import torch.fx as fx

# Framework fused ops are traced as leaf call_function nodes via the
# tracer's autowrap_functions (patched by function identity in every
# namespace during tracing).
import importlib


def _leaf_fns():
    fns = []
    for modname, names in (
            ("torchacc_amd.ops.flash_attn",
             ("flash_attn_xla", "flash_attn_varlen_xla", "flash_attn_func")),
            ("torchacc_amd.ops.rmsnorm", ("rms_norm", "fused_add_rms_norm")),
            ("torchacc_amd.ops.rope", ("apply_rotary_pos_emb",)),
            ("torchacc_amd.ops.swiglu", ("swiglu",)),
            ("torchacc_amd.ops.cross_entropy",
             ("cross_entropy", "linear_cross_entropy"))):
        mod = importlib.import_module(modname)
        for n in names:
            fns.append(getattr(mod, n))
    return tuple(fns)


class _Tracer(fx.Tracer):
    """Plain tracer + framework fused-op leaves + RMSNorm leaf modules."""

    def __init__(self):
        super().__init__(autowrap_functions=_leaf_fns())

    def is_leaf_module(self, m, qualname):
        from ..ops.linear import TunedLinear
        from ..ops.rmsnorm import RMSNorm
        if isinstance(m, (RMSNorm, TunedLinear)):
            # TunedLinear.forward branches on device/dtype (GPU hipBLASLt
            # vs F.linear) — leaf it like nn.Linear so fx never sees that
            return True
        return super().is_leaf_module(m, qualname)


def _concrete_args_from_signature(model: torch.nn.Module,
                                  input_names: List[str]):
    sig = inspect.signature(
        model.forward.__func__ if hasattr(model.forward, "__func__")
        else model.forward)
    concrete = {}
    for name, param in sig.parameters.items():
        if name in ("self",) or name in input_names:
            continue
        if param.kind in (param.VAR_POSITIONAL, param.VAR_KEYWORD):
            continue
        if param.default is inspect.Parameter.empty:
            raise ValueError(
                f"forward arg '{name}' has no default and is not in "
                f"input_names {input_names}")
        concrete[name] = param.default
    return concrete


def trace(model: torch.nn.Module,
          input_names: Optional[List[str]] = None) -> fx.GraphModule:
    """Symbolically trace ``model`` keeping only ``input_names`` as graph
    placeholders (reference trace.py:73)."""
    input_names = list(input_names or [])
    try:
        from transformers import PreTrainedModel
        is_hf = isinstance(model, PreTrainedModel)
    except ImportError:
        is_hf = False
    if is_hf:
        # transformers < 5 shipped an HF-aware tracer; transformers >= 5
        # removed transformers.utils.fx, so fall through to the generic
        # tracer (works for models whose forward branches are static given
        # the concrete defaults)
        try:
            from transformers.utils.fx import symbolic_trace as hf_st
            return hf_st(model, input_names=input_names)
        except ImportError:
            pass
    concrete = _concrete_args_from_signature(model, input_names)
    tracer = _Tracer()
    graph = tracer.trace(model, concrete_args=concrete or None)
    gm = fx.GraphModule(model, graph)
    return gm
