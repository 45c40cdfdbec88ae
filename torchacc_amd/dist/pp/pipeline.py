"""Pipeline parallelism facade (reference dist/pp/pipeline.py:13-149).

Flow: normalize split points (module objects -> qualnames) -> fx-trace if
needed -> split into per-stage submodules (pp/utils.py) -> keep only this
rank's stage (others' parameters are dropped) -> expose forward /
forward_backward through the PipeExecutor.
"""
from typing import List, Optional

import torch

from ...utils.logger import logger
from ...utils.trace import trace
from ..parallel_module import ParallelModule
from .executor import PipeExecutor
from .utils import split


def _preprocess_split_points(model, split_points) -> Optional[List[str]]:
    """Module objects/classes -> qualified names (reference :13-24)."""
    if not split_points:
        return None
    out = []
    for sp in split_points:
        if isinstance(sp, str):
            out.append(sp)
        elif isinstance(sp, torch.nn.Module):
            for name, mod in model.named_modules():
                if mod is sp:
                    out.append(name)
                    break
            else:
                raise ValueError(f"split point module not found: {sp}")
        elif isinstance(sp, type):
            matches = [
                name for name, mod in model.named_modules()
                if type(mod) is sp
            ]
            out.extend(matches)
        else:
            raise TypeError(f"unsupported split point {sp!r}")
    return out


class PipelineParallel(ParallelModule):

    def __init__(self, model: torch.nn.Module, config, **kwargs):
        super().__init__(model, config, **kwargs)
        pp_cfg = config.dist.pp
        self.num_stages = pp_cfg.size
        self.stage_id = self.mesh.get_stage_id()
        self._orig_forward = model.forward

        import torch.fx as fx
        from .hf_split import is_hf_splittable, split_hf_model
        sr = None
        if not isinstance(model, fx.GraphModule) and \
                not pp_cfg.split_points and is_hf_splittable(model):
            # HF causal LMs: split on the decoder-layer ModuleList directly
            # (transformers >= 5 removed HFTracer; reference capability
            # dist/pp/pipeline.py:38-44 restored without fx)
            sr = split_hf_model(model, self.num_stages, pp_cfg.input_names)
        if sr is None:
            if not isinstance(model, fx.GraphModule):
                try:
                    gm = trace(model, pp_cfg.input_names)
                except Exception as e:
                    raise RuntimeError(
                        "pipeline parallelism requires an fx-traceable "
                        f"model; tracing {type(model).__name__} failed "
                        f"({e}). transformers >= 5 removed its HF fx "
                        "tracer; HF causal LMs are split on their decoder "
                        "ModuleList instead — this model matches neither. "
                        "Use the native model families (torchacc_amd."
                        "models) for PP, or FSDP/DP/CP which need no "
                        "tracing") from e
            else:
                gm = model
            sp = _preprocess_split_points(model, pp_cfg.split_points)
            sr = split(gm, self.num_stages, sp)
        if self.mesh.global_rank == 0:
            for i, spec in enumerate(sr.specs):
                logger.info("stage %d: %d inputs, send %s", i,
                            len(spec.inputs), spec.send_vids)
        # keep only this rank's stage module; free the rest
        self.model = sr.submodules[self.stage_id]
        for i, sub in enumerate(sr.submodules):
            if i != self.stage_id:
                sr.submodules[i] = None  # type: ignore
        sr.submodules = [None] * self.num_stages  # executor uses self.module
        sr.submodules[self.stage_id] = self.model
        self.executor = PipeExecutor(sr, self.stage_id, self.mesh, config,
                                     self.device, self._orig_forward)
        self.model.to(self.device)

    def forward(self, *args, output_fn=None, **kwargs):
        return self.executor.forward(*args, output_fn=output_fn, **kwargs)

    def forward_backward(self, *args, output_fn=None, **kwargs):
        return self.executor.forward_backward(*args, output_fn=output_fn,
                                              **kwargs)

    def _get_underlay_model(self):
        return self.model

    def _update_underlay_model(self, model: torch.nn.Module):
        self.model = model
        self.executor.module = model
