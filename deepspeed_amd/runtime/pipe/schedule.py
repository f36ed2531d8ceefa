"""Pipeline instruction schedules (reference:
deepspeed/runtime/pipe/schedule.py — TrainSchedule.steps :197,
InferenceSchedule, instruction classes :237-376).

A schedule is a generator of instruction lists; the engine executes each
instruction via its ``_exec_*`` map. TrainSchedule is 1F1B: stage ``s`` of
``S`` runs ``min(S-s-1, M)`` warmup forwards, then alternates fwd/bwd, then
drains. In-flight activations are bounded by ``warmup+1`` buffer slots.
"""

from typing import List


class PipeInstruction:
    def __init__(self, **kwargs):
        self.kwargs = kwargs
        for k, v in kwargs.items():
            setattr(self, k, v)

    def __repr__(self):
        args = ", ".join(f"{k}={v}" for k, v in self.kwargs.items())
        return f"{type(self).__name__}({args})"


class OptimizerStep(PipeInstruction):
    pass


class ReduceGrads(PipeInstruction):
    pass


class ReduceTiedGrads(PipeInstruction):
    pass


class LoadMicroBatch(PipeInstruction):
    pass  # kwargs: buffer_id, micro_batch_id


class ForwardPass(PipeInstruction):
    pass  # kwargs: buffer_id, micro_batch_id


class BackwardPass(PipeInstruction):
    pass  # kwargs: buffer_id, micro_batch_id


class SendActivation(PipeInstruction):
    pass  # kwargs: buffer_id


class RecvActivation(PipeInstruction):
    pass  # kwargs: buffer_id


class SendGrad(PipeInstruction):
    pass  # kwargs: buffer_id


class RecvGrad(PipeInstruction):
    pass  # kwargs: buffer_id


class PipeSchedule:
    def __init__(self, micro_batches: int, stages: int, stage_id: int):
        self.micro_batches = micro_batches
        self.stages = stages
        self.stage_id = stage_id

    @property
    def is_first_stage(self):
        return self.stage_id == 0

    @property
    def is_last_stage(self):
        return self.stage_id == self.stages - 1

    def num_pipe_buffers(self) -> int:
        raise NotImplementedError

    def steps(self):
        raise NotImplementedError

    def __iter__(self):
        return self.steps()


class TrainSchedule(PipeSchedule):
    """1F1B with bounded activation memory."""

    def num_pipe_buffers(self) -> int:
        warmup = min(self.stages - self.stage_id - 1, self.micro_batches)
        return warmup + 1

    def steps(self):
        M = self.micro_batches
        warmup = min(self.stages - self.stage_id - 1, M)
        nbuf = self.num_pipe_buffers()
        fwd_mb = 0
        bwd_mb = 0

        def fwd_cmds(mb):
            buf = mb % nbuf
            cmds: List[PipeInstruction] = []
            if self.is_first_stage:
                cmds.append(LoadMicroBatch(buffer_id=buf, micro_batch_id=mb))
            else:
                cmds.append(RecvActivation(buffer_id=buf))
            if self.is_last_stage and not self.is_first_stage:
                # single-stage pipelines load ONCE: the executor fills both
                # inputs and labels from the same batch
                cmds.append(LoadMicroBatch(buffer_id=buf, micro_batch_id=mb))
            cmds.append(ForwardPass(buffer_id=buf, micro_batch_id=mb))
            if not self.is_last_stage:
                cmds.append(SendActivation(buffer_id=buf))
            return cmds

        def bwd_cmds(mb):
            buf = mb % nbuf
            cmds: List[PipeInstruction] = []
            if not self.is_last_stage:
                cmds.append(RecvGrad(buffer_id=buf))
            cmds.append(BackwardPass(buffer_id=buf, micro_batch_id=mb))
            if not self.is_first_stage:
                cmds.append(SendGrad(buffer_id=buf))
            return cmds

        for _ in range(warmup):
            yield fwd_cmds(fwd_mb)
            fwd_mb += 1
        while fwd_mb < M:
            yield fwd_cmds(fwd_mb) + bwd_cmds(bwd_mb)
            fwd_mb += 1
            bwd_mb += 1
        while bwd_mb < M:
            yield bwd_cmds(bwd_mb)
            bwd_mb += 1
        yield [ReduceTiedGrads(), ReduceGrads(), OptimizerStep()]


class InferenceSchedule(PipeSchedule):
    """Straight-through forward-only schedule."""

    def num_pipe_buffers(self) -> int:
        return 2

    def steps(self):
        nbuf = self.num_pipe_buffers()
        for mb in range(self.micro_batches):
            buf = mb % nbuf
            cmds: List[PipeInstruction] = []
            if self.is_first_stage:
                cmds.append(LoadMicroBatch(buffer_id=buf, micro_batch_id=mb))
            else:
                cmds.append(RecvActivation(buffer_id=buf))
            if self.is_last_stage and not self.is_first_stage:
                cmds.append(LoadMicroBatch(buffer_id=buf, micro_batch_id=mb))
            cmds.append(ForwardPass(buffer_id=buf, micro_batch_id=mb))
            if not self.is_last_stage:
                cmds.append(SendActivation(buffer_id=buf))
            yield cmds
