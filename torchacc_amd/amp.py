"""AMP GradScaler (reference core/amp.py:9-42).

Subclasses torch.amp.GradScaler; ``_unscale_grads_`` additionally all-reduces
the found_inf flags over the PP group so every pipeline stage skips the
optimizer step together. With the framework's fused AdamW the step is fully
SYNCFREE (reference torch_xla.amp.syncfree semantics): the found_inf flag is
handed to the optimizer kernel device-side — the update becomes a device
no-op on overflow and the host never synchronizes.
"""
import torch
import torch.distributed as dist


class GradScaler(torch.amp.GradScaler):

    def __init__(self, *args, pp_group=None, **kwargs):
        device = "cuda" if torch.cuda.is_available() else "cpu"
        if device == "cpu":
            kwargs.setdefault("enabled", False)
        super().__init__(device, *args, **kwargs)
        self._pp_group = pp_group

    def _unscale_grads_(self, optimizer, inv_scale, found_inf,
                        allow_fp16=False):
        # pure-fp16 training keeps fp32 masters inside the fused AdamW, so
        # unscaling fp16 grads directly is safe FOR THAT OPTIMIZER only;
        # torch's default forbids it because plain optimizers would update
        # fp16 params in place from fp16 grads
        from .ops.adamw import AdamW as FusedAdamW
        allow = allow_fp16 or (
            isinstance(optimizer, FusedAdamW) and
            getattr(optimizer, "use_master_weights", False))
        out = super()._unscale_grads_(optimizer, inv_scale, found_inf,
                                      allow_fp16=allow)
        if self._pp_group is not None and dist.is_initialized() and \
                dist.get_world_size(self._pp_group) > 1:
            for v in out.values():
                for t in v:
                    dist.all_reduce(t, group=self._pp_group)
        return out

    def step(self, optimizer, *args, **kwargs):
        from .ops.adamw import AdamW as FusedAdamW
        if not self._enabled or not isinstance(optimizer, FusedAdamW):
            return super().step(optimizer, *args, **kwargs)
        # syncfree path: unscale if needed, then gate in-kernel
        state = self._per_optimizer_states[id(optimizer)]
        from torch.amp.grad_scaler import OptState
        if state["stage"] is OptState.READY:
            self.unscale_(optimizer)
        found = None
        for t in state["found_inf_per_device"].values():
            found = t if found is None else found + t
        retval = optimizer.step(*args, found_inf=found, **kwargs)
        state["stage"] = OptState.STEPPED
        return retval
