"""Data parallelism: replicate the model, bucketed async grad all-reduce.

Reference semantics (dist/dp.py:12-89): post-backward coalesced all-reduce of
all grads then grad /= dp_size, with param broadcast at init. On MI355X we
register per-parameter post-accumulate-grad hooks that feed a
:class:`BucketedAllReduce` (overlapping reduction with the rest of backward
on a side HIP stream) instead of one giant coalesced op at the end.
"""
import torch
import torch.distributed as dist

from .backend import BucketedAllReduce
from .parallel_module import ParallelModule


class DataParallel(ParallelModule):

    def __init__(self, model: torch.nn.Module, config, **kwargs):
        super().__init__(model, config, **kwargs)
        self.model = model
        self.group = self.mesh.get_dp_proc_group()
        self.dp_size = self.mesh.get_dp_num()
        bucket_mb = getattr(config.dist.fsdp, "bucket_mb", 64)
        self._reducer = BucketedAllReduce(self.group, bucket_mb=bucket_mb)
        self._hooked = False
        self._pending = False
        if self.dp_size > 1:
            self._broadcast_params()
            self._install_hooks()

    def _broadcast_params(self):
        """Broadcast rank-0 params so replicas start identical
        (reference accelerate.py:142-144 broadcast_master_param)."""
        with torch.no_grad():
            for p in self.model.parameters():
                dist.broadcast(p.data, src=self._src_rank(), group=self.group)
            for b in self.model.buffers():
                if b.dtype.is_floating_point or b.dtype in (
                        torch.int64, torch.int32):
                    dist.broadcast(b.data, src=self._src_rank(),
                                   group=self.group)

    def _src_rank(self) -> int:
        groups = self.mesh.get_dp_rank_groups()
        for ranks in groups:
            if self.mesh.global_rank in ranks:
                return ranks[0]
        return 0

    def _install_hooks(self):
        for p in self.model.parameters():
            if not p.requires_grad:
                continue

            def hook(param: torch.Tensor):
                if param.grad is not None:
                    self._reducer.add(param.grad)
                    if not self._pending:
                        # finalize once, at the very end of this backward
                        self._pending = True
                        torch.autograd.Variable._execution_engine \
                            .queue_callback(self.reduce_gradients)

            p.register_post_accumulate_grad_hook(hook)
        self._hooked = True

    def forward(self, *args, **kwargs):
        return self.model(*args, **kwargs)

    def reduce_gradients(self):
        """Wait for all bucketed all-reduces and write back averaged grads.
        Call after loss.backward(), before optimizer.step()."""
        if self.dp_size > 1 and self._pending:
            self._reducer.finalize(self.dp_size)
            self._pending = False

    # optimizer.step() interposition: users of the one-call API call
    # ta.sync() (no-op) then optimizer.step(); DistributedParallel wires
    # reduce_gradients into a pre-step hook instead (see accelerate()).

    def _get_underlay_model(self):
        return self.model

    def _update_underlay_model(self, model: torch.nn.Module):
        self.model = model

    def clip_grad_norm_(self, max_grad_norm: float):
        self.reduce_gradients()
        torch.nn.utils.clip_grad_norm_(self.model.parameters(), max_grad_norm)
