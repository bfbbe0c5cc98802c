"""The Worker: interprets instruction streams against buffers, a model,
a dataset, an optimizer and the communicator grid.

Reference: shallowspeed/pipe.py:330-466.  Parity notes:
  * buffer loads with shape asserts (pipe.py:355-365),
  * PP p2p to stage±1 neighbors (pipe.py:367-381, 414-418),
  * forward maps input→output buffer (pipe.py:383-387),
  * backward_and_reduce installs DP hooks around the backward then
    resets them (pipe.py:389-400),
  * _INSTRUCTION_MAP dispatch (pipe.py:420-432),
  * PERSISTENT device buffers — the reference reallocates per batch and
    carries a TODO to persist them (pipe.py:444-445,446-454); we
    allocate once per (schedule shape) and reuse.

MI355X notes: buffers are device-resident bf16; p2p tensors go over
RCCL/xGMI; the DP all-reduce rides torch.distributed's comm stream and
overlaps backward kernels.
"""

import contextlib
import os

import torch

from ..utils import StepTimer
from . import comm as comm_mod
from .comm import GradReducer, Topology
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


class _BatchedP2P:
    """A run of >=2 consecutive p2p instructions coalesced into one
    grouped RCCL call (see Worker._coalesce_p2p)."""

    __slots__ = ("cmds",)

    def __init__(self, cmds):
        self.cmds = cmds


class Worker:
    def __init__(self, topo: Topology, model, dataset=None, optimizer=None,
                 use_dp: bool = True, bucket_bytes=None):
        self.topo = topo
        self.model = model
        self.dataset = dataset
        self.optimizer = optimizer
        self.device = getattr(model, "device", torch.device("cpu"))
        self.compute_dtype = getattr(model, "compute_dtype", torch.float32)
        self.reducer = None
        if use_dp and topo.dp > 1:
            self.reducer = GradReducer(model, topo.dp_group, bucket_bytes)
        self._in_bufs = []
        self._out_bufs = []
        self._buf_shape = None
        # async p2p bookkeeping: ("in"|"out", idx) -> pending isend work
        self._pending_send = {}
        self.instruction_times = {}  # class name -> seconds (observability)
        self.comm_stats = {}         # p2p overlap counters (debug/timing)
        self._timing = False

        self._DISPATCH = {
            ZeroGrad: self._zero_grad,
            OptimizerStep: self._optimizer_step,
            LoadMuBatchInput: self._load_input,
            LoadMuBatchTarget: self._load_target,
            RecvActivations: self._recv_activations,
            SendActivations: self._send_activations,
            RecvOutputGrad: self._recv_output_grad,
            SendInputGrad: self._send_input_grad,
            Forward: self._forward,
            BackwardGradAcc: self._backward_acc,
            BackwardGradAllReduce: self._backward_and_reduce,
        }

    # ------------------------------------------------------------ buffers
    def _ensure_buffers(self, num_buffers: int, mubatch_size: int):
        assert num_buffers % 2 == 0, "num_buffers must be even (pipe.py:446)"
        n = num_buffers // 2
        shape = (n, mubatch_size)
        if self._buf_shape == shape:
            return
        self._buf_shape = shape
        kw = dict(dtype=self.compute_dtype, device=self.device)
        # four pools: activations in/out, gradients in/out.  Separate
        # grad pools keep the forward and backward p2p streams from
        # colliding on buffer reuse in 1F1B steady state (the reference
        # conflates them, pipe.py:446-454, which only its naive/GPipe
        # orderings tolerate).
        self._in_bufs = [
            torch.zeros(mubatch_size, self.model.in_dim, **kw) for _ in range(n)
        ]
        self._out_bufs = [
            torch.zeros(mubatch_size, self.model.out_dim, **kw) for _ in range(n)
        ]
        self._gin_bufs = [
            torch.zeros(mubatch_size, self.model.out_dim, **kw) for _ in range(n)
        ]
        self._gout_bufs = [
            torch.zeros(mubatch_size, self.model.in_dim, **kw) for _ in range(n)
        ]

    # --------------------------------------------------------- execution
    def execute(self, schedule, batch_id: int):
        self._batch_id = batch_id
        # the last training stage's forward output is consumed only via
        # the head's stash (the out buffer would be dead weight: it is
        # overwritten by LoadMuBatchTarget into the grad-in pool) —
        # skip the staging copy there
        self._skip_out_copy = (
            getattr(schedule, "is_training", True)
            and self.model._training
            and self.topo.stage_id == self.topo.pp - 1
        )
        self._in_views = {}
        # deferred-µbatch wgrad: one chunked kernel per layer at the
        # optimizer step instead of a wgrad launch per µbatch
        self._defer_active = (
            (self.device.type == "cuda"
             or getattr(self, "_force_defer", False))  # CPU test hook
            and getattr(schedule, "is_training", True)
            and schedule.num_micro_batches > 1
            and hasattr(self.model, "set_defer_wgrad")
            and not getattr(self, "_force_no_defer", False)
        )
        # snapshot recv'd/loaded activations when layers may retain
        # them past the buffer's reuse: >1 µbatch in flight, or wgrad
        # deferral holding them until flush.  (Zero-copy dataset VIEWS
        # are immutable and never need the snapshot.)
        self._snapshot_inputs = (
            getattr(schedule, "max_in_flight", 2) > 1
            or self._defer_active
        )
        # µbatch-parallel HIP streams (single-stage schedules only):
        # on one stage, different µbatches' forwards/backwards are
        # data-independent (grads accumulate via f32 atomics), so their
        # per-kernel wave-ramp floors could in principle overlap on
        # side streams.  MEASURED OFF by default: a raw 2-stream
        # microbench wins (scripts/microbench_streams.py: µ4 forward
        # 133→98 µs), but the full Worker path LOSES on hardware
        # (GPipe µ4 0.355→0.431 ms, 1F1B µ8 0.60→0.96 ms — stream
        # contention on chip-filling kernels, same conclusion as the
        # round-1 side-stream wgrad experiment).  The machinery stays
        # behind SSPEED_MU_STREAMS=1 (correctness is covered by the
        # gloo GPU tests, which ran green with it forced on).
        self._mu_par = (
            os.environ.get("SSPEED_MU_STREAMS", "0") == "1"
            and self.device.type == "cuda"
            and self.topo.pp == 1
            and getattr(schedule, "is_training", True)
            and schedule.num_micro_batches > 1
            and not getattr(self, "_force_no_mu_par", False)
        )
        if self._mu_par and not hasattr(self, "_mu_streams"):
            self._mu_streams = [torch.cuda.Stream(), torch.cuda.Stream()]
        if hasattr(self.model, "set_defer_wgrad"):
            # window = schedule's in-flight bound, floored at 4:
            # GPipe (O(M) stash by design) batches all µbatches; 1F1B
            # keeps its ~warmup+1 memory bound to within a small
            # constant (≤4 retained µbatch grad/act pairs) while still
            # amortizing kernel launches 4x.  Under µbatch-parallel
            # streams the window is the whole batch: an early window
            # flush would launch on one side stream while reading
            # chunks produced on the other (cross-stream race); with
            # pp==1 there is no pipeline memory bound to protect.
            if self._mu_par:
                win = schedule.num_micro_batches
            else:
                win = max(4, getattr(schedule, "max_in_flight",
                                     schedule.num_micro_batches))
            self.model.set_defer_wgrad(self._defer_active, window=win)
        mub = self.dataset.mubatch_size if self.dataset is not None else \
            self._buf_shape[1]
        self._ensure_buffers(schedule.num_buffers, mub)
        self._mu_fork()
        batch_ok = self.topo.is_distributed and \
            comm_mod.can_batch_p2p(self.device)
        for commands in schedule.steps():
            if batch_ok:
                commands = self._coalesce_p2p(commands)
            for cmd in commands:
                if isinstance(cmd, _BatchedP2P):
                    self._run_batched_p2p(cmd)
                    continue
                fn = self._DISPATCH[type(cmd)]
                if self._timing:
                    with StepTimer(self.device) as t:
                        fn(cmd)
                    key = type(cmd).__name__
                    self.instruction_times[key] = (
                        self.instruction_times.get(key, 0.0) + t.elapsed
                    )
                else:
                    fn(cmd)
        self.flush_sends()

    # ------------------------------------------- batched p2p (RCCL path)
    # Runs of >=2 consecutive p2p instructions within one schedule step
    # (e.g. send-activations-to-next + recv-grad-from-next — the same
    # peer, both directions) become ONE grouped ncclSend/Recv call via
    # dist.batch_isend_irecv: NCCL's matched-order group semantics make
    # the bidirectional edge robust regardless of per-call enqueue
    # interleaving across ranks.  Single p2p instructions keep the
    # (cheaper) per-op path.
    _P2P_TYPES = (SendActivations, SendInputGrad,
                  RecvActivations, RecvOutputGrad)

    def _coalesce_p2p(self, commands):
        out, run = [], []
        for cmd in commands:
            if isinstance(cmd, self._P2P_TYPES):
                run.append(cmd)
                continue
            if run:
                out.append(_BatchedP2P(run) if len(run) > 1 else run[0])
                run = []
            out.append(cmd)
        if run:
            out.append(_BatchedP2P(run) if len(run) > 1 else run[0])
        return out

    def _run_batched_p2p(self, batch):
        ops = []
        for cmd in batch.cmds:
            if isinstance(cmd, SendActivations):
                self._wait_buffer("out", cmd.buffer_idx)
                ops.append(("send", self._out_bufs[cmd.buffer_idx],
                            self.topo.next_rank))
            elif isinstance(cmd, SendInputGrad):
                self._wait_buffer("gout", cmd.buffer_idx)
                ops.append(("send", self._gout_bufs[cmd.buffer_idx],
                            self.topo.prev_rank))
            elif isinstance(cmd, RecvActivations):
                self._in_views.pop(cmd.buffer_idx, None)
                self._wait_buffer("in", cmd.buffer_idx)
                ops.append(("recv", self._in_bufs[cmd.buffer_idx],
                            self.topo.prev_rank))
            else:  # RecvOutputGrad
                self._wait_buffer("gin", cmd.buffer_idx)
                ops.append(("recv", self._gin_bufs[cmd.buffer_idx],
                            self.topo.next_rank))
        for w in comm_mod.batch_p2p(ops):
            # NCCL: stream-level wait (compute after this sees the
            # recv'd data; the send buffers become reusable in stream
            # order — no handle parking needed for batched sends)
            w.wait()

    def enable_instruction_timing(self, on: bool = True):
        self._timing = on

    # -------------------------------------------------- hipGraph replay
    # The per-step kernel sequence is launch-bound at MLP scale
    # (~20 small kernels/step); capturing it in a hipGraph collapses
    # per-kernel host launch + boundary gaps into one replay
    # (MI355X price list: eager host launch ≈3.3-3.8 µs/kernel vs
    # graph replay ≈10-16 µs per whole step).
    #
    # Replay needs fixed addresses, so the batch's rows are first
    # staged into persistent buffers OUTSIDE the graph, and the
    # captured Load instructions read from the staging buffers.
    def execute_graphed(self, schedule, batch_id: int):
        if self.device.type != "cuda" or self.topo.world > 1:
            return self.execute(schedule, batch_id)
        ds = self.dataset
        lb = ds.local_batch_size
        if not hasattr(self, "_graphs"):
            self._graphs = {}
        # few batches: capture one graph PER batch_id (loads bake the
        # batch's device addresses in — no staging copy per step).
        # many batches: one graph reading persistent staging buffers,
        # refilled with a single D2D copy per step.
        per_batch = ds.num_batches() <= 8
        key = (type(schedule).__name__, schedule.num_micro_batches,
               schedule.num_stages, schedule.stage_id, ds.mubatch_size,
               batch_id if per_batch else -1)
        if not per_batch:
            if not hasattr(self, "_staged_x"):
                self._staged_x = torch.empty_like(ds.x_compute[:lb])
                self._staged_y = torch.empty_like(ds.y_compute[:lb])
            b0 = batch_id * lb
            self._staged_x.copy_(ds.x_compute[b0:b0 + lb], non_blocking=True)
            self._staged_y.copy_(ds.y_compute[b0:b0 + lb], non_blocking=True)
        g = self._graphs.get(key)
        if g is None:
            self._use_staged = not per_batch
            try:
                # eager warmup (allocates buffers, builds SGD desc) on a
                # side stream, then capture
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    self.execute(schedule, batch_id)
                torch.cuda.current_stream().wait_stream(s)
                torch.cuda.synchronize()
                g = torch.cuda.CUDAGraph()
                with torch.cuda.graph(g):
                    self.execute(schedule, batch_id)
            except Exception:
                self._graphs[key] = False
                return self.execute(schedule, batch_id)
            finally:
                self._use_staged = False
            self._graphs[key] = g
        elif g is False:  # capture failed before: stay eager
            return self.execute(schedule, batch_id)
        g.replay()

    # ----------------------------------------------------- async p2p
    # Sends are nonblocking isends (RCCL p2p over the direct xGMI link
    # between stage-adjacent GPUs); a buffer is only waited on when it
    # is about to be OVERWRITTEN.  This is the double-buffered async
    # p2p the reference leaves as a TODO (pipe.py:269-272) and what
    # makes the 1F1B steady state deadlock-free with blocking recvs.
    def _wait_buffer(self, kind, idx):
        h = self._pending_send.pop((kind, idx), None)
        if h is not None:
            # overlap observability (the comm/compute race guard — the
            # analog of the reference's by-construction Iallreduce
            # safety note at pipe.py:313): count how often the wait
            # actually blocked vs the send having already drained
            if self._timing:
                try:
                    done = h.is_completed()
                except Exception:
                    done = False
                k = "p2p_wait_blocked" if not done else "p2p_wait_free"
                self.comm_stats[k] = self.comm_stats.get(k, 0) + 1
            h.wait()

    def _isend(self, kind, idx, tensor, dst):
        self._wait_buffer(kind, idx)
        self._pending_send[(kind, idx)] = comm_mod.isend_tensor(tensor, dst)

    def flush_sends(self):
        for h in self._pending_send.values():
            h.wait()
        self._pending_send.clear()

    # ------------------------------------------- µbatch stream plumbing
    # Strict fork/join phase discipline: side streams always WAIT the
    # main stream at fork, main WAITS the side streams at join (before
    # the AllReduce backward and the optimizer step).  Every
    # cross-stream tensor handoff (stashes, deferred wgrad chunks read
    # by the main-stream flush) crosses a join, and allocator reuse of
    # side-allocated blocks happens only after the next fork — so the
    # stream-aware caching allocator needs no record_stream calls.
    def _mu_ctx(self, mubatch_id):
        if not getattr(self, "_mu_par", False):
            return contextlib.nullcontext()
        return torch.cuda.stream(self._mu_streams[mubatch_id % 2])

    def _mu_fork(self):
        if getattr(self, "_mu_par", False):
            cur = torch.cuda.current_stream()
            for s in self._mu_streams:
                s.wait_stream(cur)

    def _mu_join(self):
        if getattr(self, "_mu_par", False):
            cur = torch.cuda.current_stream()
            for s in self._mu_streams:
                cur.wait_stream(s)

    # ------------------------------------------------- instruction impls
    def _zero_grad(self, cmd):
        self.model.zero_grad()
        self._mu_fork()

    def _optimizer_step(self, cmd):
        self._mu_join()
        if getattr(self, "_defer_active", False):
            # normally a no-op: the AllReduce backward flushed every
            # layer mid-backward (_flush_in_backward).  Leftovers mean
            # the schedule never issued BackwardGradAllReduce — under
            # DP that would desync replicas, so fail loudly.
            n = self.model.flush_wgrads()
            if n and self.reducer is not None:
                raise RuntimeError(
                    f"{n} layers' deferred wgrads flushed AFTER the DP "
                    f"reduction — schedule issued no BackwardGradAllReduce")
        self.optimizer.step()
        if hasattr(self.model, "invalidate_fp8"):
            self.model.invalidate_fp8()

    def _load_input(self, cmd):
        if getattr(self, "_use_staged", False):
            mb = self.dataset.mubatch_size
            x = self._staged_x[cmd.mubatch_id * mb:(cmd.mubatch_id + 1) * mb]
        else:
            x = self.dataset.micro_batch_input(self._batch_id, cmd.mubatch_id)
        buf = self._in_bufs[cmd.buffer_idx]
        assert x.shape == buf.shape, (x.shape, buf.shape)  # pipe.py:357-360
        if x.dtype == self.compute_dtype and x.device == buf.device:
            # zero-copy: the forward reads the dataset slice directly
            # (the staging buffer exists for the RECV path; stage 0
            # never receives).  The slice is read-only to the model.
            self._in_views[cmd.buffer_idx] = x
            return
        self._wait_buffer("in", cmd.buffer_idx)
        self._in_views.pop(cmd.buffer_idx, None)
        with self._mu_ctx(cmd.mubatch_id):
            buf.copy_(x.to(self.compute_dtype), non_blocking=True)

    def _load_target(self, cmd):
        if getattr(self, "_use_staged", False):
            mb = self.dataset.mubatch_size
            y = self._staged_y[cmd.mubatch_id * mb:(cmd.mubatch_id + 1) * mb]
        else:
            y = self.dataset.micro_batch_target(self._batch_id, cmd.mubatch_id)
        self._wait_buffer("gin", cmd.buffer_idx)
        buf = self._gin_bufs[cmd.buffer_idx]
        assert y.shape == buf.shape, (y.shape, buf.shape)  # pipe.py:362-365
        with self._mu_ctx(cmd.mubatch_id):
            buf.copy_(y.to(self.compute_dtype), non_blocking=True)

    def _recv_activations(self, cmd):
        self._in_views.pop(cmd.buffer_idx, None)
        self._wait_buffer("in", cmd.buffer_idx)
        comm_mod.recv_tensor(self._in_bufs[cmd.buffer_idx], self.topo.prev_rank)

    def _send_activations(self, cmd):
        self._isend("out", cmd.buffer_idx,
                    self._out_bufs[cmd.buffer_idx], self.topo.next_rank)

    def _recv_output_grad(self, cmd):
        self._wait_buffer("gin", cmd.buffer_idx)
        comm_mod.recv_tensor(self._gin_bufs[cmd.buffer_idx], self.topo.next_rank)

    def _send_input_grad(self, cmd):
        self._isend("gout", cmd.buffer_idx,
                    self._gout_bufs[cmd.buffer_idx], self.topo.prev_rank)

    def _forward(self, cmd):
        x = self._in_views.get(cmd.in_buffer)
        from_view = x is not None
        if x is None:
            x = self._in_bufs[cmd.in_buffer]
        with self._mu_ctx(cmd.mubatch_id):
            if self.model._training and self._snapshot_inputs and not from_view:
                # Layers stash their input per µbatch for wgrad; the
                # input buffer is SHARED across µbatches (overwritten by
                # the next Load/Recv), so snapshot it when more than one
                # µbatch can be in flight.  (The reference stashes the
                # live buffer reference — layers.py:117 + pipe.py:447-454
                # — which is only safe for its naive schedule ordering.)
                x = x.clone()
            y = self.model.forward(x, cmd.mubatch_id)
            if getattr(self, "_skip_out_copy", False):
                return
            self._wait_buffer("out", cmd.out_buffer)
            self._out_bufs[cmd.out_buffer].copy_(y)

    def _backward_acc(self, cmd, _mu_stream_ok=True):
        with (self._mu_ctx(cmd.mubatch_id) if _mu_stream_ok
              else contextlib.nullcontext()):
            g = self._gin_bufs[cmd.out_buffer]
            if getattr(self, "_defer_active", False) and \
                    self.topo.stage_id != self.topo.pp - 1:
                # non-last stages hand the RECV'D grad buffer to the
                # stage's last Linear, whose deferred wgrad would retain
                # the reference past the buffer's next overwrite —
                # snapshot it.  (The last stage's gin holds the TARGET,
                # which the loss head consumes immediately.)
                g = g.clone()
            d = self.model.backward(g, cmd.mubatch_id)
            # stage 0 never sends input grads, so skip the staging copy
            if d is not None and self.topo.stage_id != 0:
                self._wait_buffer("gout", cmd.in_buffer)
                self._gout_bufs[cmd.in_buffer].copy_(d)
            return d

    def _backward_and_reduce(self, cmd):
        """Install DP hooks, run backward, reset hooks — the one
        'inversion' in the stack (pipe.py:389-400 ↔ layers.py:201-213):
        grad-ready → bucket all-reduce launches mid-backward.

        Deferred-wgrad mode: every training schedule places
        BackwardGradAllReduce on the temporally LAST backward of the
        batch (naive: last µbatch; GPipe: µbatch 0, processed last;
        1F1B: last cooldown backward), so _flush_in_backward makes each
        layer's grads final the moment that layer's backward completes
        — its bucket's all-reduce then overlaps the remaining layers'
        dgrad/wgrad kernels, same as the eager path."""
        # the AllReduce backward is the batch's synchronization point:
        # join the µbatch side streams (its flush reads every µbatch's
        # deferred chunks) and run on the MAIN stream.
        self._mu_join()
        if getattr(self, "_defer_active", False):
            self.model._flush_in_backward = True
        if self.reducer is not None:
            self.reducer.reset()
            self.model.register_grad_hook(self.reducer.param_done)
            self.model.register_post_grad_hook(
                lambda params: self.reducer.finalize())
        try:
            self._backward_acc(cmd, _mu_stream_ok=False)
        finally:
            self.model._flush_in_backward = False
            if self.reducer is not None:
                self.model.reset_grad_hooks()
                self.model.reset_post_grad_hooks()
