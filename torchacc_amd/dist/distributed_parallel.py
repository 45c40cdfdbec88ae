"""Strategy composer (reference dist/distributed_parallel.py:11-111).

Wrap order: PP -> FSDP -> DP. DP is wrapped only when FSDP is absent (when
both are requested, the FSDP engine itself performs the hybrid-shard DP
all-reduce after its reduce-scatter, reference fsdp.py:196-216).
TP rewrites happen before wrapping (module surgery in dist/tp.py).
"""
import torch

from .parallel_module import ParallelModule


class DistributedParallel(ParallelModule):

    def __init__(self, model: torch.nn.Module, config, **kwargs):
        super().__init__(model, config, **kwargs)
        self.pp_wrapper = None
        self.fsdp_wrapper = None
        self.dp_wrapper = None
        m = model
        if self.has_pp:
            from .pp.pipeline import PipelineParallel
            self.pp_wrapper = PipelineParallel(m, config, **kwargs)
            m = self.pp_wrapper
        if self.has_fsdp or config.dist.fsdp.wrap_layer_cls:
            from .fsdp import FullyShardedDataParallel
            if self.pp_wrapper is not None:
                inner = self.pp_wrapper._get_underlay_model()
                fsdp = FullyShardedDataParallel(inner, config, **kwargs)
                self.pp_wrapper._update_underlay_model(fsdp)
                self.fsdp_wrapper = fsdp
            else:
                self.fsdp_wrapper = FullyShardedDataParallel(
                    m, config, **kwargs)
                m = self.fsdp_wrapper
        elif self.has_dp:
            from .dp import DataParallel
            self.dp_wrapper = DataParallel(m, config, **kwargs)
            m = self.dp_wrapper
        self.module = m

    def forward(self, *args, output_fn=None, **kwargs):
        out = self.module(*args, **kwargs)
        if output_fn is not None:
            out = output_fn(out)
        return out

    def forward_backward(self, *args, **kwargs):
        if self.pp_wrapper is None:
            raise RuntimeError(
                "forward_backward requires pipeline parallelism (pp.size>1); "
                "use forward() + loss.backward() otherwise")
        return self.pp_wrapper.forward_backward(*args, **kwargs)

    def _get_underlay_model(self):
        m = self.module
        while isinstance(m, ParallelModule):
            m = m._get_underlay_model()
        return m

    def _update_underlay_model(self, model: torch.nn.Module):
        if isinstance(self.module, ParallelModule):
            self.module._update_underlay_model(model)
        else:
            self.module = model

    def clip_grad_norm_(self, max_grad_norm: float, norm_type: float = 2.0):
        if self.fsdp_wrapper is not None:
            return self.fsdp_wrapper.clip_grad_norm_(max_grad_norm,
                                                     norm_type)
        if self.dp_wrapper is not None:
            return self.dp_wrapper.clip_grad_norm_(max_grad_norm)
        return torch.nn.utils.clip_grad_norm_(self.parameters(),
                                              max_grad_norm,
                                              norm_type=norm_type)

    # FSDP optim-state passthroughs (reference distributed_parallel.py:63-111)
    def no_sync(self):
        """Gradient-accumulation context: skip grad reduction inside; the
        first synchronized backward after the context reduces the
        accumulated total (FSDP engine; no-op without FSDP)."""
        if self.fsdp_wrapper is not None:
            return self.fsdp_wrapper.no_sync()
        import contextlib
        return contextlib.nullcontext()

    def sharded_state_dict(self):
        assert self.fsdp_wrapper is not None
        return self.fsdp_wrapper.sharded_state_dict()

    def full_state_dict(self):
        assert self.fsdp_wrapper is not None
        return self.fsdp_wrapper.full_state_dict()

    def sharded_optim_state_dict(self, optim):
        from .state_dict_utils import sharded_optim_state_dict
        return sharded_optim_state_dict(self.fsdp_wrapper, optim)

    def full_optim_state_dict(self, optim):
        from .state_dict_utils import full_optim_state_dict
        return full_optim_state_dict(self.fsdp_wrapper, optim)

    def optim_state_dict_to_load(self, state, optim):
        from .state_dict_utils import optim_state_dict_to_load
        return optim_state_dict_to_load(self.fsdp_wrapper, state, optim)
