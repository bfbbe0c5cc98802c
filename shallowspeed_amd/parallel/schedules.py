"""Pipeline schedules as pure-data instruction-stream generators.

Reference: shallowspeed/pipe.py:141-299 — Schedule ABC + Naive, GPipe,
Inference, and a PipeDream stub (pipe.py:297-299, NotImplementedError).
Here PipeDream-Flush (1F1B) is implemented for real.

All schedules are pure data: steps() yields lists of instructions and
never touches a communicator, so schedule properties (ordering,
happens-before, pairwise send/recv consistency across stages) are
testable single-process — the reference's own test TODO
(tests/test_schedules.py:4-10) asks for exactly that strengthening.
"""

from .instructions import (
    BackwardGradAcc,
    BackwardGradAllReduce,
    Forward,
    LoadMuBatchInput,
    LoadMuBatchTarget,
    OptimizerStep,
    RecvActivations,
    RecvOutputGrad,
    SendActivations,
    SendInputGrad,
    ZeroGrad,
)


class Schedule:
    """Reference: pipe.py:141-181."""

    def __init__(self, num_micro_batches: int, num_stages: int, stage_id: int):
        assert num_micro_batches >= 1
        assert 0 <= stage_id < num_stages
        self.num_micro_batches = num_micro_batches
        self.num_stages = num_stages
        self.stage_id = stage_id

    def steps(self):
        raise NotImplementedError

    @property
    def num_buffers(self) -> int:
        """Must be even: input+output buffer pairs (pipe.py:156-164)."""
        raise NotImplementedError

    # predicates — pipe.py:166-181
    @property
    def is_first_stage(self):
        return self.stage_id == 0

    @property
    def is_last_stage(self):
        return self.stage_id == self.num_stages - 1

    def is_first_mubatch(self, mubatch_id):
        return mubatch_id == 0

    def is_last_mubatch(self, mubatch_id):
        return mubatch_id == self.num_micro_batches - 1

    # training schedules run backward; InferenceSchedule overrides.
    is_training = True

    # buffer assignment: µbatch m round-robins over the buffer pairs.
    # Sends are ASYNC (the Worker isends and waits only when a buffer
    # is about to be overwritten), so schedules that keep >1 µbatch in
    # flight use num_buffers=4 (two in/out pairs) — the double-buffered
    # p2p the reference left as a TODO (pipe.py:269-272).
    def _buf(self, m):
        return m % (self.num_buffers // 2)

    @property
    def max_in_flight(self) -> int:
        """Upper bound on µbatches with a live activation stash at any
        point — 1 means the Worker may pass the shared input buffer to
        forward without snapshotting it."""
        return self.num_micro_batches

    # shared fragments ------------------------------------------------
    def _acquire_input(self, m):
        if self.is_first_stage:
            return [LoadMuBatchInput(m, self._buf(m))]
        return [RecvActivations(m, self._buf(m))]

    def _acquire_output_grad(self, m):
        if self.is_last_stage:
            return [LoadMuBatchTarget(m, self._buf(m))]
        return [RecvOutputGrad(m, self._buf(m))]

    def _backward(self, m, allreduce):
        cls = BackwardGradAllReduce if allreduce else BackwardGradAcc
        return [cls(m, self._buf(m), self._buf(m))]


class NaiveParallelSchedule(Schedule):
    """One µbatch fully forward+backward before the next.
    Reference: pipe.py:184-222.  The DP all-reduce-interleaved backward
    variant lands on the LAST µbatch (pipe.py:209-212)."""

    def steps(self):
        yield [ZeroGrad()]
        for m in range(self.num_micro_batches):
            b = self._buf(m)
            cmds = []
            cmds += self._acquire_input(m)
            cmds += [Forward(m, b, b)]
            if not self.is_last_stage:
                cmds += [SendActivations(m, b)]
            cmds += self._acquire_output_grad(m)
            cmds += self._backward(m, allreduce=self.is_last_mubatch(m))
            if not self.is_first_stage:
                cmds += [SendInputGrad(m, b)]
            yield cmds
        yield [OptimizerStep()]

    @property
    def num_buffers(self):
        return 2

    @property
    def max_in_flight(self):
        return 1  # fwd+bwd complete before the next µbatch starts


class GPipeSchedule(Schedule):
    """All forwards in order, then all backwards in REVERSE order, so
    the all-reduce-interleaved backward lands on µbatch 0 (the last one
    processed).  Reference: pipe.py:225-272 (reverse order
    pipe.py:233-235, allreduce-on-first-µbatch pipe.py:246-248, last
    stage discards its forward output instead of sending,
    pipe.py:262-266)."""

    def steps(self):
        yield [ZeroGrad()]
        for m in range(self.num_micro_batches):
            b = self._buf(m)
            cmds = []
            cmds += self._acquire_input(m)
            cmds += [Forward(m, b, b)]
            if not self.is_last_stage:
                cmds += [SendActivations(m, b)]
            yield cmds
        for m in reversed(range(self.num_micro_batches)):
            b = self._buf(m)
            cmds = []
            cmds += self._acquire_output_grad(m)
            cmds += self._backward(m, allreduce=self.is_first_mubatch(m))
            if not self.is_first_stage:
                cmds += [SendInputGrad(m, b)]
            yield cmds
        yield [OptimizerStep()]

    @property
    def num_buffers(self):
        # two in/out pairs: async sends overlap the next µbatch's
        # compute (fixes reference TODO pipe.py:269-272)
        return 4


class PipeDreamFlushSchedule(Schedule):
    """PipeDream-Flush / 1F1B — REAL implementation (the reference
    declares it and raises NotImplementedError, pipe.py:297-299;
    registered in its CLI at train.py:50-54).

    Stage s of P runs `warmup = min(M, P-1-s)` forwards, then a steady
    1F1B phase alternating one forward with one backward, then a
    cooldown of the remaining backwards.  Within a batch all µbatches
    use the same weights (flush variant — no weight versioning needed);
    peak activation stash is `warmup+1` µbatches instead of GPipe's M,
    which is what bounds memory for deep pipelines.

    The all-reduce-interleaved backward is the LAST backward (µbatch
    M-1), mirroring the naive schedule's convention.
    """

    def steps(self):
        P, M, s = self.num_stages, self.num_micro_batches, self.stage_id
        warmup = min(M, P - 1 - s)
        yield [ZeroGrad()]
        # warmup forwards
        for m in range(warmup):
            cmds = []
            cmds += self._acquire_input(m)
            cmds += [Forward(m, self._buf(m), self._buf(m))]
            if not self.is_last_stage:
                cmds += [SendActivations(m, self._buf(m))]
            yield cmds
        # steady 1F1B
        for m in range(warmup, M):
            b = m - warmup
            cmds = []
            cmds += self._acquire_input(m)
            cmds += [Forward(m, self._buf(m), self._buf(m))]
            if not self.is_last_stage:
                cmds += [SendActivations(m, self._buf(m))]
            cmds += self._acquire_output_grad(b)
            cmds += self._backward(b, allreduce=self.is_last_mubatch(b))
            if not self.is_first_stage:
                cmds += [SendInputGrad(b, self._buf(b))]
            yield cmds
        # cooldown backwards
        for b in range(M - warmup, M):
            cmds = []
            cmds += self._acquire_output_grad(b)
            cmds += self._backward(b, allreduce=self.is_last_mubatch(b))
            if not self.is_first_stage:
                cmds += [SendInputGrad(b, self._buf(b))]
            yield cmds
        yield [OptimizerStep()]

    @property
    def num_buffers(self):
        # two in/out pairs — enough for 1F1B's one-forward-one-backward
        # steady state with async sends
        return 4

    @property
    def max_in_flight(self):
        return min(self.num_micro_batches,
                   self.num_stages - self.stage_id) # warmup+1


class InferenceSchedule(Schedule):
    """Forward-only pipeline for eval.  Reference: pipe.py:275-294
    (used by compute_accuracy, train.py:32-37)."""

    is_training = False

    def steps(self):
        for m in range(self.num_micro_batches):
            b = self._buf(m)
            cmds = []
            cmds += self._acquire_input(m)
            cmds += [Forward(m, b, b)]
            if not self.is_last_stage:
                cmds += [SendActivations(m, b)]
            yield cmds

    @property
    def num_buffers(self):
        return 2

    @property
    def max_in_flight(self):
        return 1  # eval stashes nothing


SCHEDULES = {
    "naive": NaiveParallelSchedule,
    "gpipe": GPipeSchedule,
    "pipedream": PipeDreamFlushSchedule,
    "1f1b": PipeDreamFlushSchedule,
}
