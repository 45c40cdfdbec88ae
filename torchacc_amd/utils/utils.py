"""Generic helpers: partitioning, pytree walkers, module lookup.

Reimplements the semantics of the reference's torchacc/utils/utils.py:15-373
(call_to_str, partition_uniform/partition_balanced, get_module_class_from_name,
recursively_apply/apply_to_tensors, convert_outputs_to_fp32) with a fresh,
eager-ROCm-oriented implementation.
"""
from bisect import bisect_left
from typing import Any, Callable, List, Optional, Sequence

import torch


def call_to_str(base: str, *args, **kwargs) -> str:
    """Render ``base(args..., kw=...)`` for log messages."""
    parts = [repr(a) for a in args]
    parts += [f"{k}={v!r}" for k, v in kwargs.items()]
    return f"{base}({', '.join(parts)})"


def partition_uniform(num_items: int, num_parts: int) -> List[int]:
    """Split ``num_items`` into ``num_parts`` contiguous ranges as evenly as
    possible. Returns ``num_parts + 1`` boundaries (first 0, last num_items).
    """
    if num_parts <= 0:
        raise ValueError("num_parts must be positive")
    base = num_items // num_parts
    rem = num_items % num_parts
    bounds = [0]
    for p in range(num_parts):
        bounds.append(bounds[-1] + base + (1 if p < rem else 0))
    return bounds


def _prefix_sums(weights: Sequence[float]) -> List[float]:
    out = [0.0]
    for w in weights:
        out.append(out[-1] + float(w))
    return out


def partition_balanced(weights: Sequence[float], num_parts: int) -> List[int]:
    """Balanced contiguous partition of weighted items: minimize the max part
    weight (binary search over the bottleneck, like DeepSpeed's ds_utils).
    Returns ``num_parts + 1`` boundaries.
    """
    n = len(weights)
    if num_parts <= 0:
        raise ValueError("num_parts must be positive")
    if n < num_parts:
        raise ValueError(f"cannot split {n} items into {num_parts} parts")
    prefix = _prefix_sums(weights)

    def can_split(limit: float) -> Optional[List[int]]:
        bounds = [0]
        cur = 0
        for _ in range(num_parts):
            # furthest j with sum(weights[cur:j]) <= limit
            target = prefix[cur] + limit
            j = bisect_left(prefix, target, lo=cur + 1)
            if j <= len(prefix) - 1 and abs(prefix[j] - target) < 1e-12:
                pass  # exact hit, keep j
            elif j > cur + 1:
                j -= 1
            if j <= cur:
                return None
            # must leave at least one item per remaining part
            remaining_parts = num_parts - len(bounds)
            j = min(j, n - remaining_parts)
            if j <= cur:
                return None
            bounds.append(j)
            cur = j
        return bounds if bounds[-1] == n else None

    lo = max(float(w) for w in weights)
    hi = prefix[-1]
    best = can_split(hi)
    for _ in range(64):
        mid = (lo + hi) / 2
        b = can_split(mid)
        if b is not None:
            best, hi = b, mid
        else:
            lo = mid
    assert best is not None
    return best


def get_module_class_from_name(module: torch.nn.Module,
                               name: str) -> Optional[type]:
    """Find a submodule class by its (unqualified) class name."""
    if module.__class__.__name__ == name:
        return module.__class__
    for child in module.children():
        cls = get_module_class_from_name(child, name)
        if cls is not None:
            return cls
    return None


def recursively_apply(func: Callable,
                      data: Any,
                      *args,
                      test_type: Callable[[Any], bool] = lambda t: isinstance(
                          t, torch.Tensor),
                      error_on_other_type: bool = False,
                      **kwargs) -> Any:
    """Apply ``func`` to every element of nested list/tuple/dict structures
    that passes ``test_type`` (tensors by default)."""
    if isinstance(data, (list, tuple)):
        mapped = [
            recursively_apply(
                func,
                o,
                *args,
                test_type=test_type,
                error_on_other_type=error_on_other_type,
                **kwargs) for o in data
        ]
        return type(data)(mapped) if not isinstance(data, tuple) or not hasattr(
            data, "_fields") else type(data)(*mapped)
    if isinstance(data, dict):
        return type(data)({
            k: recursively_apply(
                func,
                v,
                *args,
                test_type=test_type,
                error_on_other_type=error_on_other_type,
                **kwargs) for k, v in data.items()
        })
    if test_type(data):
        return func(data, *args, **kwargs)
    if error_on_other_type:
        raise TypeError(f"Unsupported type {type(data)}")
    return data


def apply_to_tensors(func: Callable, data: Any) -> Any:
    return recursively_apply(func, data)


def convert_outputs_to_fp32(func: Callable) -> Callable:
    """Wrap a forward so floating outputs are upcast to fp32 (used after
    autocast regions, reference fsdp.py:169-180)."""

    def _cast(t: torch.Tensor) -> torch.Tensor:
        if t.is_floating_point() and t.dtype in (torch.float16,
                                                 torch.bfloat16):
            return t.float()
        return t

    def wrapper(*args, **kwargs):
        return recursively_apply(_cast, func(*args, **kwargs))

    return wrapper
