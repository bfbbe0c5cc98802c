"""The schedule ISA: pure-data instruction dataclasses.

Reference: shallowspeed/pipe.py:12-138 — 12 dataclasses which a
Schedule emits and a Worker interprets.  Kept pure data so any
(num_stages, stage_id, num_µbatches) combination can be instantiated
and property-tested in a single process (reference test strategy,
tests/test_schedules.py).

Extensions over the reference:
  * every buffer-touching instruction carries BOTH a mubatch_id and an
    explicit buffer index (the reference hardwires buffer 0 since its
    schedules only ever use num_buffers=2) — needed by PipeDream-Flush
    and by async double-buffered p2p.
"""

from dataclasses import dataclass


@dataclass(frozen=True)
class PipeInstr:
    pass


@dataclass(frozen=True)
class ZeroGrad(PipeInstr):
    """Zero the stage's flat grad buffer.  Reference: pipe.py:16-23."""


@dataclass(frozen=True)
class OptimizerStep(PipeInstr):
    """Fused SGD over all stage params.  Reference: pipe.py:26-32."""


@dataclass(frozen=True)
class LoadMuBatchInput(PipeInstr):
    """Dataset µbatch inputs → input buffer (first stage only).
    Reference: pipe.py:118-127."""
    mubatch_id: int
    buffer_idx: int = 0


@dataclass(frozen=True)
class LoadMuBatchTarget(PipeInstr):
    """Dataset µbatch targets → output buffer (last stage only).
    Reference: pipe.py:129-138."""
    mubatch_id: int
    buffer_idx: int = 0


@dataclass(frozen=True)
class RecvActivations(PipeInstr):
    """P2P recv activations from stage-1 into input buffer.
    Reference: pipe.py:35-45."""
    mubatch_id: int
    buffer_idx: int = 0


@dataclass(frozen=True)
class SendActivations(PipeInstr):
    """P2P send output buffer to stage+1.  Reference: pipe.py:47-56."""
    mubatch_id: int
    buffer_idx: int = 0


@dataclass(frozen=True)
class RecvOutputGrad(PipeInstr):
    """P2P recv output-grad from stage+1 into output buffer.
    Reference: pipe.py:58-67."""
    mubatch_id: int
    buffer_idx: int = 0


@dataclass(frozen=True)
class SendInputGrad(PipeInstr):
    """P2P send input buffer (dL/dinput) to stage-1.
    Reference: pipe.py:69-77."""
    mubatch_id: int
    buffer_idx: int = 0


@dataclass(frozen=True)
class Forward(PipeInstr):
    """model.forward(input buffer) → output buffer.
    Reference: pipe.py:86-93."""
    mubatch_id: int
    in_buffer: int = 0
    out_buffer: int = 0


@dataclass(frozen=True)
class BackwardGradAcc(PipeInstr):
    """model.backward(output buffer) → input buffer, grads ACCUMULATE.
    Reference: pipe.py:96-104."""
    mubatch_id: int
    in_buffer: int = 0
    out_buffer: int = 0


@dataclass(frozen=True)
class BackwardGradAllReduce(PipeInstr):
    """Same as BackwardGradAcc but with the DP bucketed all-reduce
    hooks installed — as each bucket's last layer finishes backward its
    async RCCL all-reduce launches, overlapping the remaining backward
    compute.  Reference: pipe.py:107-115 (+ hooks pipe.py:302-327)."""
    mubatch_id: int
    in_buffer: int = 0
    out_buffer: int = 0
