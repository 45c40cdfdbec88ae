"""Context-parallel group construction (reference init_group.py:20-112).

``initialize_context_parallel(cp_size, intra_size)`` arranges the world as a
``[dp, inter, intra]`` grid (outer -> inner rank stride) and creates:

- the INTRA group (adjacent ranks, xGMI-local): Ulysses head-scatter
  all-to-all — bandwidth-heavy, benefits from the 7 xGMI links;
- the INTER group (strided ranks, cross-node): ring attention — latency
  tolerant, overlapped P2P;
- the full CONTEXT group (all cp ranks of this dp replica).
"""
from typing import Optional

import torch.distributed as dist

_INTRA_CP_GROUP = None
_INTER_CP_GROUP = None
_CONTEXT_PARALLEL_GROUP = None
_CP_SIZES = None


def initialize_context_parallel(cp_size: int, intra_size: Optional[int] = None):
    global _INTRA_CP_GROUP, _INTER_CP_GROUP, _CONTEXT_PARALLEL_GROUP, _CP_SIZES
    if _CONTEXT_PARALLEL_GROUP is not None:
        return
    assert dist.is_initialized()
    world = dist.get_world_size()
    rank = dist.get_rank()
    intra_size = intra_size or cp_size
    assert cp_size % intra_size == 0
    inter_size = cp_size // intra_size
    assert world % cp_size == 0
    dp = world // cp_size
    _CP_SIZES = (cp_size, inter_size, intra_size)

    for d in range(dp):
        base = d * cp_size
        # full CP group
        ranks = list(range(base, base + cp_size))
        g = dist.new_group(ranks=ranks)
        if rank in ranks:
            _CONTEXT_PARALLEL_GROUP = g
        # intra groups: contiguous runs of intra_size
        for i in range(inter_size):
            ranks = [base + i * intra_size + j for j in range(intra_size)]
            g = dist.new_group(ranks=ranks)
            if rank in ranks and intra_size > 1:
                _INTRA_CP_GROUP = g
        # inter groups: stride intra_size
        for j in range(intra_size):
            ranks = [base + i * intra_size + j for i in range(inter_size)]
            g = dist.new_group(ranks=ranks)
            if rank in ranks and inter_size > 1:
                _INTER_CP_GROUP = g


def get_intra_cp_group():
    return _INTRA_CP_GROUP


def get_inter_cp_group():
    return _INTER_CP_GROUP


def get_context_parallel_group():
    return _CONTEXT_PARALLEL_GROUP


def get_cp_sizes():
    return _CP_SIZES


def destroy_context_parallel():
    global _INTRA_CP_GROUP, _INTER_CP_GROUP, _CONTEXT_PARALLEL_GROUP, _CP_SIZES
    _INTRA_CP_GROUP = None
    _INTER_CP_GROUP = None
    _CONTEXT_PARALLEL_GROUP = None
    _CP_SIZES = None
