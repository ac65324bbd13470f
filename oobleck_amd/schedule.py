# 1F1B pipeline schedule — a self-contained restatement of the command
# stream the reference emits (OobleckPipelineSchedule,
# /root/reference/oobleck/execution/pipeline.py:24-84) including the
# deepspeed TrainSchedule helpers the reference calls but does not define
# (_step_to_micro_batch / _valid_micro_batch / _buffer_idx /
# num_pipe_buffers / prev_stage / next_stage — SURVEY.md §5 "third-party
# runtime code").  The mapping formulas are re-derived from the classic 1F1B
# timetable: stage s runs F_m at half-step t = s + 2m and B_m at
# t = 2*stages - 1 - s + 2m; even stages do forward on even t, odd stages on
# odd t.  Behaviour is pinned by tests/test_schedule.py's cross-stage
# simulation (send/recv pairing, 1F1B ordering invariants).
from __future__ import annotations

from dataclasses import dataclass


@dataclass(frozen=True)
class PipeInstruction:
    buffer_id: int

    @property
    def kwargs(self) -> dict:
        return {"buffer_id": self.buffer_id}


class LoadMicroBatch(PipeInstruction): pass
class ForwardPass(PipeInstruction): pass
class BackwardPass(PipeInstruction): pass
class SendActivation(PipeInstruction): pass
class RecvActivation(PipeInstruction): pass
class SendGrad(PipeInstruction): pass
class RecvGrad(PipeInstruction): pass
class OptimizerStep(PipeInstruction): pass  # never emitted (pipeline.py:80)


class OobleckPipelineSchedule:
    """Emits, per half-step, the same command multiset in the same order as
    the reference's steps() (pipeline.py:34-84): p2p exchanges first
    (SendGrad/RecvActivation on forward steps, RecvGrad/SendActivation on
    backward steps), then LoadMicroBatch on the first/last stage, then the
    compute command.  Allreduce + optimizer step are deliberately excluded
    (the engine drives them per training step, engine.py:645-649)."""

    def __init__(self, micro_batches: int, stages: int, stage_id: int,
                 min_pipe_buffers: int = 2):
        # min_pipe_buffers > 2 deepens the slot rotation (used by the pp1
        # fwd/bwd overlap so the forward stream can run further ahead);
        # the command SEQUENCE is unchanged, only the buffer assignment.
        self.min_pipe_buffers = min_pipe_buffers
        assert 0 <= stage_id < stages
        self.micro_batches = micro_batches
        self.stages = stages
        self.stage_id = stage_id

    # --- deepspeed TrainSchedule helper semantics (restated) ---------------
    @property
    def prev_stage(self) -> int:
        return self.stage_id - 1

    @property
    def next_stage(self) -> int:
        return self.stage_id + 1

    def _valid_stage(self, stage_id: int) -> bool:
        return 0 <= stage_id < self.stages

    def _valid_micro_batch(self, micro_batch_id: int) -> bool:
        return 0 <= micro_batch_id < self.micro_batches

    def num_pipe_buffers(self) -> int:
        buffers = min(self.stages - self.stage_id, self.micro_batches)
        return max(getattr(self, "min_pipe_buffers", 2), buffers)

    def _buffer_idx(self, micro_batch_id: int) -> int:
        assert self._valid_micro_batch(micro_batch_id)
        return micro_batch_id % self.num_pipe_buffers()

    def _step_to_micro_batch(self, step_id: int) -> tuple[int, bool]:
        s, S = self.stage_id, self.stages
        even_step, even_stage = step_id % 2 == 0, s % 2 == 0
        if even_step == even_stage:
            # forward:  t = s + 2m  ->  m = (t - s) / 2
            base = step_id // 2 if even_step else (step_id - 1) // 2
            return base - s // 2, True
        # backward:  t = 2S - 1 - s + 2m  ->  m = (t + 1 + s) / 2 - S
        if even_step:  # odd stage
            return step_id // 2 - S + (s + 1) // 2, False
        return (step_id - 1) // 2 - S + 1 + s // 2, False

    # --- the command stream (reference steps(), pipeline.py:34-84) ---------
    def steps(self):
        prev_micro_batch_id = -1
        total_steps = 2 * (self.micro_batches + self.stages - 1)
        prev_buffer = curr_buffer = None
        for step_id in range(total_steps):
            micro_batch_id, is_forward = self._step_to_micro_batch(step_id)
            if self._valid_micro_batch(prev_micro_batch_id):
                prev_buffer = self._buffer_idx(prev_micro_batch_id)
            if self._valid_micro_batch(micro_batch_id):
                curr_buffer = self._buffer_idx(micro_batch_id)

            cmds: list[PipeInstruction] = []
            if is_forward:
                if (self._valid_micro_batch(prev_micro_batch_id)
                        and self._valid_stage(self.prev_stage)):
                    cmds.append(SendGrad(prev_buffer))
                if (self._valid_micro_batch(micro_batch_id)
                        and self._valid_stage(self.prev_stage)):
                    cmds.append(RecvActivation(curr_buffer))
            else:
                if (self._valid_micro_batch(micro_batch_id)
                        and self._valid_stage(self.next_stage)):
                    cmds.append(RecvGrad(curr_buffer))
                if (self._valid_micro_batch(prev_micro_batch_id)
                        and self._valid_stage(self.next_stage)):
                    cmds.append(SendActivation(prev_buffer))

            if self.stage_id == 0 or self.stage_id == self.stages - 1:
                if is_forward and self._valid_micro_batch(micro_batch_id):
                    cmds.append(LoadMicroBatch(curr_buffer))

            if self._valid_micro_batch(micro_batch_id):
                cmds.append(ForwardPass(curr_buffer) if is_forward
                            else BackwardPass(curr_buffer))

            prev_micro_batch_id = micro_batch_id
            yield cmds

    def __iter__(self):
        return self.steps()
