"""PipeDream-Flush (1F1B) instruction schedules
(reference dist/pp/schedule.py:10-409, DeepSpeed-style).

Instructions are interpreted by the executor; generation here is pure.
Warmup forwards per stage = min(micro_batches, stages - stage_id - 1),
steady state = alternating 1F1B, cooldown drains the remaining backwards.
"""
from dataclasses import dataclass
from typing import Iterator, List


@dataclass(frozen=True)
class PipeInstruction:
    micro_batch: int


class LoadMicroBatch(PipeInstruction):
    pass


class ForwardPass(PipeInstruction):
    pass


class BackwardPass(PipeInstruction):
    pass


class SendActivation(PipeInstruction):
    pass


class RecvActivation(PipeInstruction):
    pass


class SendGrad(PipeInstruction):
    pass


class RecvGrad(PipeInstruction):
    pass


class ReduceGrads(PipeInstruction):
    pass


class OptimizerStep(PipeInstruction):
    pass


class PipeSchedule:
    def __init__(self, micro_batches: int, stages: int, stage_id: int):
        self.micro_batches = micro_batches
        self.stages = stages
        self.stage_id = stage_id
        self.is_first = stage_id == 0
        self.is_last = stage_id == stages - 1

    def steps(self) -> Iterator[List[PipeInstruction]]:
        raise NotImplementedError


class PipeDreamFlushInfer(PipeSchedule):
    """Forward-only (reference :122-153)."""

    def steps(self):
        for mb in range(self.micro_batches):
            cmds: List[PipeInstruction] = []
            if self.is_first:
                cmds.append(LoadMicroBatch(mb))
            else:
                cmds.append(RecvActivation(mb))
                cmds.append(LoadMicroBatch(mb))
            cmds.append(ForwardPass(mb))
            if not self.is_last:
                cmds.append(SendActivation(mb))
            yield cmds


class PipeDreamFlushTrain(PipeSchedule):
    """1F1B with flush (reference :156-227)."""

    def steps(self):
        M, S, s = self.micro_batches, self.stages, self.stage_id
        warmup = min(M, S - s - 1)
        steady = M - warmup
        # warmup forwards
        for mb in range(warmup):
            cmds: List[PipeInstruction] = []
            if not self.is_first:
                cmds.append(RecvActivation(mb))
            cmds.append(LoadMicroBatch(mb))
            cmds.append(ForwardPass(mb))
            if not self.is_last:
                cmds.append(SendActivation(mb))
            yield cmds
        # steady 1F1B
        for i in range(steady):
            fwd_mb = warmup + i
            bwd_mb = i
            cmds = []
            if not self.is_first:
                cmds.append(RecvActivation(fwd_mb))
            cmds.append(LoadMicroBatch(fwd_mb))
            cmds.append(ForwardPass(fwd_mb))
            if not self.is_last:
                cmds.append(SendActivation(fwd_mb))
                cmds.append(RecvGrad(bwd_mb))
            cmds.append(BackwardPass(bwd_mb))
            if not self.is_first:
                cmds.append(SendGrad(bwd_mb))
            yield cmds
        # cooldown backwards
        for mb in range(steady, M):
            cmds = []
            if not self.is_last:
                cmds.append(RecvGrad(mb))
            cmds.append(BackwardPass(mb))
            if not self.is_first:
                cmds.append(SendGrad(mb))
            yield cmds
        yield [ReduceGrads(0)]


def create_scheduler(kind: str, micro_batches: int, stages: int,
                     stage_id: int) -> PipeSchedule:
    if kind == "train":
        return PipeDreamFlushTrain(micro_batches, stages, stage_id)
    if kind == "infer":
        return PipeDreamFlushInfer(micro_batches, stages, stage_id)
    raise ValueError(kind)
