"""SPMD-FSDP compatibility wrapper (reference dist/spmd_fsdp.py:20-95).

The reference's SPMD path wrapped the model in torch_xla FSDPv2 over a GSPMD
mesh and let the compiler insert collectives. The eager MI355X backend has a
single sharding engine (the flat-param FSDP in dist/fsdp.py), so
``use_spmd=True`` maps onto it: same sharding semantics (ZeRO-3 over the
fsdp axis), with ``shard_output_callable`` applied to forward outputs for
API compatibility.
"""
import torch

from .fsdp import FullyShardedDataParallel


class SpmdFullyShardedDataParallel(FullyShardedDataParallel):

    def __init__(self, model: torch.nn.Module, config, **kwargs):
        super().__init__(model, config, **kwargs)
        self._shard_output = config.dist.fsdp.shard_output_callable

    def forward(self, *args, **kwargs):
        out = super().forward(*args, **kwargs)
        if self._shard_output is not None:
            out = self._shard_output(out)
        return out
