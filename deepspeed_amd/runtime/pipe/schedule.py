"""Pipeline schedules as instruction streams.

Parity: reference `runtime/pipe/schedule.py:189` (TrainSchedule 1F1B),
`:135` (InferenceSchedule), instruction IR @327-489.
"""
from ..utils import call_to_str


class PipeInstruction:
    def __init__(self, **kwargs):
        self.kwargs = kwargs
        for k, v in kwargs.items():
            setattr(self, k, v)

    def __repr__(self):
        return call_to_str(self.__class__.__name__, **self.kwargs)


class OptimizerStep(PipeInstruction):
    pass


class ReduceGrads(PipeInstruction):
    pass


class BufferOpInstruction(PipeInstruction):
    def __init__(self, buffer_id, **kwargs):
        super().__init__(buffer_id=buffer_id, **kwargs)


class LoadMicroBatch(BufferOpInstruction):
    pass


class ForwardPass(BufferOpInstruction):
    pass


class BackwardPass(BufferOpInstruction):
    pass


class SendActivation(BufferOpInstruction):
    pass


class RecvActivation(BufferOpInstruction):
    pass


class SendGrad(BufferOpInstruction):
    pass


class RecvGrad(BufferOpInstruction):
    pass


class PipeSchedule:
    """Base: yields lists of instructions per step."""

    def __init__(self, micro_batches, stages, stage_id):
        self.micro_batches = micro_batches
        self.stages = stages
        self.stage_id = stage_id

    @property
    def is_first_stage(self):
        return self.stage_id == 0

    @property
    def is_last_stage(self):
        return self.stage_id == self.stages - 1

    def steps(self):
        raise NotImplementedError

    def __iter__(self):
        return iter(self.steps())


class TrainSchedule(PipeSchedule):
    """1F1B: warmup forwards, steady fwd/bwd interleave, cooldown backwards,
    then grad reduce + optimizer step."""

    def steps(self):
        M = self.micro_batches
        warmup = min(self.stages - self.stage_id - 1, M)
        cmds = []

        def fwd(m):
            step = []
            if self.is_first_stage:
                step.append(LoadMicroBatch(m))
            else:
                step.append(RecvActivation(m))
            if self.is_last_stage:
                step.append(LoadMicroBatch(m))  # labels
            step.append(ForwardPass(m))
            if not self.is_last_stage:
                step.append(SendActivation(m))
            return step

        def bwd(b):
            step = []
            if not self.is_last_stage:
                step.append(RecvGrad(b))
            step.append(BackwardPass(b))
            if not self.is_first_stage:
                step.append(SendGrad(b))
            return step

        for m in range(warmup):
            cmds.append(fwd(m))
        for m in range(warmup, M):
            cmds.append(fwd(m))
            cmds.append(bwd(m - warmup))
        for b in range(M - warmup, M):
            cmds.append(bwd(b))
        cmds.append([ReduceGrads(), OptimizerStep()])
        return cmds


class InferenceSchedule(PipeSchedule):
    def steps(self):
        cmds = []
        for m in range(self.micro_batches):
            step = []
            if self.is_first_stage:
                step.append(LoadMicroBatch(m))
            else:
                step.append(RecvActivation(m))
            step.append(ForwardPass(m))
            if not self.is_last_stage:
                step.append(SendActivation(m))
            cmds.append(step)
        return cmds
