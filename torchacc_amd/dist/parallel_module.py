"""Base class for all parallel wrappers (reference dist/parallel_module.py:8-69)."""
from abc import abstractmethod

import torch

from .. import dist as ta_dist


class ParallelModule(torch.nn.Module):
    """Holds the config, device and mesh, exposes strategy flags, and defines
    the underlay-model plumbing each wrapper implements."""

    def __init__(self, model: torch.nn.Module, config, **kwargs):
        super().__init__()
        self.config = config
        self.mesh = config.get_mesh()
        if torch.cuda.is_available():
            self.device = torch.device("cuda", ta_dist.local_rank())
        else:
            self.device = torch.device("cpu")
        d = config.dist
        self.has_dp = d.dp.size > 1
        self.has_tp = d.tp.size > 1
        self.has_pp = d.pp.size > 1
        self.has_fsdp = d.fsdp.size > 1
        self.spmd_fsdp = d.fsdp.use_spmd

    @abstractmethod
    def _get_underlay_model(self) -> torch.nn.Module:
        ...

    @abstractmethod
    def _update_underlay_model(self, model: torch.nn.Module) -> None:
        ...

    def clip_grad_norm_(self, max_grad_norm: float):
        torch.nn.utils.clip_grad_norm_(self.parameters(), max_grad_norm)

    def forward_backward(self, *args, **kwargs):
        raise NotImplementedError(
            "forward_backward is only provided by pipeline-parallel wrappers")
