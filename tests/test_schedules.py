"""Schedule property tests — schedules are pure data, inspected without
running them (reference tests/test_schedules.py:29-102), PLUS the
stronger happens-before / deadlock-freedom checks the reference's own
TODO asks for (tests/test_schedules.py:4-10): we simulate rendezvous
send/recv across all stages of a pipeline and assert progress."""

import itertools

import pytest

from shallowspeed_amd.parallel.instructions import (
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
from shallowspeed_amd.parallel.schedules import (
    GPipeSchedule,
    InferenceSchedule,
    NaiveParallelSchedule,
    PipeDreamFlushSchedule,
)

TRAIN_SCHEDS = [NaiveParallelSchedule, GPipeSchedule, PipeDreamFlushSchedule]


def flat(sched):
    return [c for step in sched.steps() for c in step]


@pytest.mark.parametrize("cls", TRAIN_SCHEDS)
@pytest.mark.parametrize("stages,stage", [(1, 0), (2, 0), (2, 1), (4, 1), (4, 3)])
@pytest.mark.parametrize("m", [1, 4])
def test_basic_structure(cls, stages, stage, m):
    cmds = flat(cls(m, stages, stage))
    # ZeroGrad exactly once, first (reference :29-40)
    assert isinstance(cmds[0], ZeroGrad)
    assert sum(isinstance(c, ZeroGrad) for c in cmds) == 1
    # OptimizerStep exactly once, last
    assert isinstance(cmds[-1], OptimizerStep)
    assert sum(isinstance(c, OptimizerStep) for c in cmds) == 1
    # exactly one all-reduce backward and it is the LAST backward
    bwds = [c for c in cmds if isinstance(c, (BackwardGradAcc, BackwardGradAllReduce))]
    assert len(bwds) == m
    assert isinstance(bwds[-1], BackwardGradAllReduce)
    assert sum(isinstance(c, BackwardGradAllReduce) for c in cmds) == 1
    # every µbatch forwarded and backwarded exactly once
    fwd_ids = sorted(c.mubatch_id for c in cmds if isinstance(c, Forward))
    bwd_ids = sorted(c.mubatch_id for c in bwds)
    assert fwd_ids == list(range(m)) and bwd_ids == list(range(m))
    # F(m) happens before B(m)
    for mm in range(m):
        fi = next(i for i, c in enumerate(cmds)
                  if isinstance(c, Forward) and c.mubatch_id == mm)
        bi = next(i for i, c in enumerate(cmds)
                  if isinstance(c, (BackwardGradAcc, BackwardGradAllReduce))
                  and c.mubatch_id == mm)
        assert fi < bi


@pytest.mark.parametrize("cls", TRAIN_SCHEDS)
def test_stage_role_io(cls):
    first = flat(cls(4, 3, 0))
    mid = flat(cls(4, 3, 1))
    last = flat(cls(4, 3, 2))
    # first stage loads inputs, never targets, never recvs acts (ref :43-69)
    assert any(isinstance(c, LoadMuBatchInput) for c in first)
    assert not any(isinstance(c, (LoadMuBatchTarget, RecvActivations,
                                  SendInputGrad)) for c in first)
    # mid stage recv/send both directions, loads nothing
    for t in (RecvActivations, SendActivations, RecvOutputGrad, SendInputGrad):
        assert any(isinstance(c, t) for c in mid)
    assert not any(isinstance(c, (LoadMuBatchInput, LoadMuBatchTarget))
                   for c in mid)
    # last stage loads targets, never sends acts
    assert any(isinstance(c, LoadMuBatchTarget) for c in last)
    assert not any(isinstance(c, (SendActivations, RecvOutputGrad))
                   for c in last)


def test_gpipe_all_fwd_before_bwd_reverse_order():
    cmds = flat(GPipeSchedule(4, 2, 0))
    last_f = max(i for i, c in enumerate(cmds) if isinstance(c, Forward))
    first_b = min(i for i, c in enumerate(cmds)
                  if isinstance(c, (BackwardGradAcc, BackwardGradAllReduce)))
    assert last_f < first_b
    bwd_ids = [c.mubatch_id for c in cmds
               if isinstance(c, (BackwardGradAcc, BackwardGradAllReduce))]
    assert bwd_ids == [3, 2, 1, 0]  # reverse (reference pipe.py:233-235)
    # allreduce variant lands on µbatch 0 (pipe.py:246-248)
    ar = [c for c in cmds if isinstance(c, BackwardGradAllReduce)]
    assert ar[0].mubatch_id == 0


def test_naive_interleaved():
    cmds = flat(NaiveParallelSchedule(3, 1, 0))
    seq = [(type(c).__name__, getattr(c, "mubatch_id", None))
           for c in cmds if isinstance(c, (Forward, BackwardGradAcc,
                                           BackwardGradAllReduce))]
    assert seq == [("Forward", 0), ("BackwardGradAcc", 0),
                   ("Forward", 1), ("BackwardGradAcc", 1),
                   ("Forward", 2), ("BackwardGradAllReduce", 2)]


def test_1f1b_warmup_depth_and_memory_bound():
    """Stage s runs min(M, P-1-s) warmup forwards; at most warmup+1
    activations in flight (the 1F1B memory bound)."""
    P, M = 4, 8
    for s in range(P):
        cmds = flat(PipeDreamFlushSchedule(M, P, s))
        in_flight, peak = 0, 0
        for c in cmds:
            if isinstance(c, Forward):
                in_flight += 1
                peak = max(peak, in_flight)
            elif isinstance(c, (BackwardGradAcc, BackwardGradAllReduce)):
                in_flight -= 1
        warmup = min(M, P - 1 - s)
        assert peak == warmup + 1, (s, peak, warmup)


def test_inference_schedule_fwd_only():
    cmds = flat(InferenceSchedule(2, 2, 0))
    assert all(not isinstance(c, (BackwardGradAcc, BackwardGradAllReduce,
                                  ZeroGrad, OptimizerStep)) for c in cmds)
    assert sum(isinstance(c, Forward) for c in cmds) == 2


# ---------------------------------------------------------------------
# rendezvous simulator: deadlock-freedom + matched channel order
# ---------------------------------------------------------------------

def simulate_pipeline(cls, M, P):
    """Run all P stages' instruction streams under the Worker's ACTUAL
    comm semantics — async isend (buffer freed only when the peer's
    matching recv consumes it), blocking recv, and wait-before-
    overwrite on shared buffers.  Returns the per-channel message log;
    raises on deadlock or on a channel-order mismatch (a send whose
    head-of-queue recv is not its pair)."""
    progs = [flat(cls(M, P, s)) for s in range(P)]
    pcs = [0] * P
    log = []
    # directed channel (src,dst) -> FIFO of (send_cmd, (kind, idx))
    chan = {}
    # per-stage set of buffers with an unconsumed outbound send
    pending = [set() for _ in range(P)]

    pair = {SendActivations: RecvActivations, SendInputGrad: RecvOutputGrad}

    def writes(c):
        if isinstance(c, (LoadMuBatchInput, RecvActivations)):
            return ("in", c.buffer_idx)
        if isinstance(c, (LoadMuBatchTarget, RecvOutputGrad)):
            return ("gin", c.buffer_idx)
        if isinstance(c, Forward):
            return ("out", c.out_buffer)
        if isinstance(c, (BackwardGradAcc, BackwardGradAllReduce)):
            return ("gout", c.in_buffer)
        return None

    total = sum(len(p) for p in progs)
    done = 0
    while done < total:
        progressed = False
        for s in range(P):
            while pcs[s] < len(progs[s]):
                c = progs[s][pcs[s]]
                w = writes(c)
                # wait-before-overwrite: blocked while an outbound send
                # from this buffer is unconsumed
                if w is not None and w in pending[s]:
                    break
                if isinstance(c, (RecvActivations, RecvOutputGrad)):
                    src = s - 1 if isinstance(c, RecvActivations) else s + 1
                    q = chan.get((src, s), [])
                    if not q:
                        break  # blocking recv, nothing sent yet
                    send_cmd, src_buf = q.pop(0)
                    assert type(c) is pair[type(send_cmd)] and \
                        c.mubatch_id == send_cmd.mubatch_id, (
                            f"channel order mismatch: {send_cmd} vs {c}")
                    pending[src].discard(src_buf)
                    log.append((src, s, type(c).__name__, c.mubatch_id))
                elif isinstance(c, (SendActivations, SendInputGrad)):
                    dst = s + 1 if isinstance(c, SendActivations) else s - 1
                    kind = "out" if isinstance(c, SendActivations) else "gout"
                    chan.setdefault((s, dst), []).append((c, (kind, c.buffer_idx)))
                    pending[s].add((kind, c.buffer_idx))
                pcs[s] += 1
                done += 1
                progressed = True
        if not progressed:
            stuck = [(s, progs[s][pcs[s]]) for s in range(P)
                     if pcs[s] < len(progs[s])]
            raise AssertionError(f"deadlock: {stuck}")
    assert all(not q for q in chan.values()), "unconsumed messages"
    return log


@pytest.mark.parametrize("cls", TRAIN_SCHEDS + [InferenceSchedule])
@pytest.mark.parametrize("P", [2, 3, 4])
@pytest.mark.parametrize("M", [1, 2, 4, 8])
def test_no_deadlock_under_rendezvous(cls, P, M):
    simulate_pipeline(cls, M, P)


@pytest.mark.parametrize("P,M", itertools.product([2, 4], [4, 8]))
def test_1f1b_activation_order(P, M):
    """Activations for µbatch m cross edge (s,s+1) in µbatch order."""
    log = simulate_pipeline(PipeDreamFlushSchedule, M, P)
    for s in range(P - 1):
        acts = [mb for (src, dst, kind, mb) in log
                if kind == "RecvActivations" and src == s]
        assert acts == list(range(M))
        grads = [mb for (src, dst, kind, mb) in log
                 if kind == "RecvOutputGrad" and dst == s]
        assert grads == list(range(M))
