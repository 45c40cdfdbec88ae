"""Pipeline parallelism facade (reference dist/pp/pipeline.py:13-149).

Round-1 scaffold: full fx-split + 1F1B executor lands with the PP milestone.
"""
import torch

from ..parallel_module import ParallelModule


class PipelineParallel(ParallelModule):

    def __init__(self, model: torch.nn.Module, config, **kwargs):
        super().__init__(model, config, **kwargs)
        raise NotImplementedError(
            "pipeline parallelism is not wired up yet in this build")

    def _get_underlay_model(self):
        return self.model

    def _update_underlay_model(self, model):
        self.model = model
