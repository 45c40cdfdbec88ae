"""AMP GradScaler (reference core/amp.py:9-42).

Subclasses torch.amp.GradScaler; ``_unscale_grads_`` additionally all-reduces
the found_inf flags over the PP group so every pipeline stage skips the
optimizer step together. The flag stays device-side (no host sync) and can be
fed to ops.AdamW's syncfree ``found_inf`` gate.
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
        out = super()._unscale_grads_(optimizer, inv_scale, found_inf,
                                      allow_fp16)
        if self._pp_group is not None and dist.is_initialized() and \
                dist.get_world_size(self._pp_group) > 1:
            for v in out.values():
                for t in v:
                    dist.all_reduce(t, group=self._pp_group)
        return out
