"""Micro-batch splitting (reference dist/pp/microbatch.py:7-48)."""
import inspect
from typing import Dict, List

import torch


def bind_args_to_kwargs(forward, args, kwargs) -> Dict:
    """Bind positional args to the original forward signature's names."""
    sig = inspect.signature(forward)
    bound = sig.bind_partial(*args, **kwargs)
    return dict(bound.arguments)


def split_microbatches(kwargs: Dict, num_micro_batches: int) -> List[Dict]:
    """Chunk every tensor along dim 0 into num_micro_batches pieces;
    non-tensors are replicated."""
    chunked = {}
    for k, v in kwargs.items():
        if isinstance(v, torch.Tensor):
            # uneven chunks would be over-weighted: the executor scales each
            # micro-batch loss by 1/num_micro_batches
            assert v.shape[0] % num_micro_batches == 0, \
                (f"batch dim {v.shape[0]} of '{k}' must be divisible by "
                 f"{num_micro_batches} micro-batches")
            chunked[k] = list(v.chunk(num_micro_batches, dim=0))
        else:
            chunked[k] = [v] * num_micro_batches
    return [
        {k: v[i] for k, v in chunked.items()}
        for i in range(num_micro_batches)
    ]
