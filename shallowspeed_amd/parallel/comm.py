"""Communicator grid + bucketed gradient reduction over RCCL/xGMI.

Reference: the reference builds two orthogonal MPI communicators from
one world with a PP-MAJOR rank layout (train.py:87-94): consecutive
ranks form one pipeline, dp_comm = Split(rank % PP), pp_comm =
Split(rank // PP).  Per-parameter nonblocking in-place Iallreduce is
fired from grad hooks (pipe.py:302-327) — its own docstring notes that
per-param messages are wasteful and real DDP buckets (pipe.py:309-310).

MI355X-native design:
  * torch.distributed process groups; backend "nccl" IS RCCL on ROCm,
    collectives ride xGMI (7 p2p links/GPU); "gloo" is used for
    CPU-only multi-process tests.
  * gradients live in ONE flat f32 buffer per stage (layers.py
    materialize_device); the GradReducer slices it into buckets along
    layer boundaries in BACKWARD order and fires one async all-reduce
    per bucket the moment its last parameter's grad is ready — the
    collective overlaps the remaining backward compute.  On xGMI the
    small-model regime is latency-bound, so the default bucket is
    large enough that a whole MNIST-scale stage is one bucket.
  * PP p2p uses dist.send/recv on the device tensors (RCCL p2p over
    the direct xGMI link between stage-adjacent GPUs).
"""

import datetime
import os
from dataclasses import dataclass, field
from typing import Optional

import torch
import torch.distributed as dist


@dataclass
class Topology:
    """Who am I in the DP×PP grid (PP-major, reference train.py:87-94)."""

    rank: int = 0
    world: int = 1
    dp: int = 1
    pp: int = 1
    tp: int = 1
    dp_group: Optional[object] = None
    pp_group: Optional[object] = None
    tp_group: Optional[object] = None
    device: torch.device = field(default_factory=lambda: torch.device("cpu"))

    @property
    def stage_id(self):
        return (self.rank // self.tp) % self.pp

    @property
    def pipeline_id(self):
        return self.rank // self.pp

    @property
    def tp_rank(self):
        # TP is the INNERMOST axis: consecutive ranks form one TP
        # group (xGMI-adjacent on a standard node layout), DP strides
        # across groups
        return self.rank % self.tp

    @property
    def dp_rank(self):
        # modulo dp so TP topologies read the SAME data shard on every
        # rank of a TP group (the batch is replicated across TP); with
        # tp=1 this is the original PP-major rank // pp
        return (self.rank // (self.pp * self.tp)) % self.dp

    @property
    def prev_rank(self):
        """Global rank of stage-1 in my pipeline (pipe.py:414-418)."""
        return self.rank - 1

    @property
    def next_rank(self):
        return self.rank + 1

    @property
    def is_distributed(self):
        return self.world > 1


def init_topology(dp: int, pp: int, backend: Optional[str] = None,
                  device: Optional[torch.device] = None,
                  tp: int = 1) -> Topology:
    """Initialize the process group and the DP/PP subgroup grid (or,
    with tp>1, a pure tensor-parallel group over the whole world —
    TP composes with dp=pp=1 in this round).

    Single-process (dp=pp=1, no env rendezvous) returns a trivial
    topology without touching torch.distributed.
    """
    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    if tp > 1:
        assert pp == 1, "TP composes with DP (pp=1) this round"
        assert dp * tp == world, f"DP({dp})×TP({tp}) != world({world})"
    else:
        assert dp * pp == world, f"DP({dp})×PP({pp}) != world({world})"

    if device is None:
        if torch.cuda.is_available():
            local = int(os.environ.get("LOCAL_RANK", rank))
            device = torch.device("cuda", local % torch.cuda.device_count())
        else:
            device = torch.device("cpu")
    if device.type == "cuda":
        if device.index is None:
            local = int(os.environ.get("LOCAL_RANK", rank))
            device = torch.device("cuda", local % torch.cuda.device_count())
        torch.cuda.set_device(device)

    if world == 1:
        return Topology(device=device)

    if backend is None:
        backend = "nccl" if device.type == "cuda" else "gloo"
    if not dist.is_initialized():
        # Bounded timeout: a mismatched collective should FAIL the run
        # loudly within minutes, not hang it until the driver's clock
        # kills the box.
        timeout = datetime.timedelta(
            seconds=int(os.environ.get("SSPEED_DIST_TIMEOUT", "240")))
        kw = {"backend": backend, "timeout": timeout}
        if backend == "nccl" and device.type == "cuda":
            # Binding the device at init lets torch create the RCCL
            # communicator EAGERLY (and derive subgroups via
            # ncclCommSplit) instead of lazily on the first collective,
            # where cross-rank first-use ordering can deadlock.
            kw["device_id"] = device
        dist.init_process_group(**kw)

    if tp > 1:
        # DP×TP grid, TP innermost: tp group d = [d*tp .. d*tp+tp),
        # dp group t = [t, t+tp, t+2tp, ...).  EVERY rank calls
        # new_group for EVERY group in the same order.
        tp_groups = {}
        for d in range(dp):
            tp_groups[d] = dist.new_group([d * tp + t for t in range(tp)])
        dp_groups = {}
        for t in range(tp):
            dp_groups[t] = dist.new_group([d * tp + t for d in range(dp)])
        topo = Topology(rank=rank, world=world, dp=dp, pp=1, tp=tp,
                        dp_group=dp_groups[rank % tp] if dp > 1 else None,
                        tp_group=tp_groups[rank // tp], device=device)
        # deterministic-order eager communicator creation (disjoint
        # groups per phase, world barriers between phases)
        dev = device if device.type == "cuda" else torch.device("cpu")
        t1 = torch.ones(1, device=dev)
        _barrier(topo)
        if dp > 1:
            dist.all_reduce(t1, group=topo.dp_group)
        _barrier(topo)
        dist.all_reduce(t1, group=topo.tp_group)
        _barrier(topo)
        return topo

    # orthogonal subgroups; EVERY rank must call new_group for EVERY
    # group in the same order.
    dp_groups = {}
    for stage in range(pp):
        ranks = [stage + pipe * pp for pipe in range(dp)]
        dp_groups[stage] = dist.new_group(ranks) if world > 1 else None
    pp_groups = {}
    for pipe in range(dp):
        ranks = [pipe * pp + s for s in range(pp)]
        pp_groups[pipe] = dist.new_group(ranks) if world > 1 else None

    topo = Topology(
        rank=rank, world=world, dp=dp, pp=pp,
        dp_group=dp_groups[rank % pp],
        pp_group=pp_groups[rank // pp],
        device=device,
    )
    _warmup_communicators(topo)
    if device.type == "cuda":
        _probe_xgmi_adjacency(topo)
    return topo


def _barrier(topo):
    if topo.device.type == "cuda" and dist.get_backend() == "nccl":
        dist.barrier(device_ids=[topo.device.index])
    else:
        dist.barrier()


def _warmup_communicators(topo):
    """Force every communicator to initialize NOW, in an order that is
    identical across ranks.

    RCCL/NCCL communicators are created lazily on first use; if two
    subgroups' first collectives interleave differently across ranks the
    lazy init can deadlock.  The dp groups are pairwise disjoint (one
    per stage), as are the pp groups (one per pipeline), so each phase
    below is race-free, and the world barriers order the phases.
    First hardware contact with RCCL happens HERE, at init, where a
    failure is attributable — not mid-schedule.
    """
    dev = topo.device if topo.device.type == "cuda" else torch.device("cpu")
    t = torch.ones(1, device=dev)
    _barrier(topo)                                   # world comm
    if topo.dp > 1 and topo.dp_group is not None:
        dist.all_reduce(t, group=topo.dp_group)      # all dp groups (disjoint)
    _barrier(topo)
    if topo.pp > 1 and topo.pp_group is not None:
        dist.all_reduce(t, group=topo.pp_group)      # all pp groups (disjoint)
    _barrier(topo)


def _probe_xgmi_adjacency(topo):
    """Log whether PP stage edges map to peer-accessible (xGMI) device
    pairs — the topology sanity probe SURVEY §2.4 #2 asks for (the
    reference's own nod is the Split_type(TYPE_SOCKET) comment at
    train.py:90-91).  On an 8-GPU MI355X node the xGMI fabric is fully
    connected (7 links/GPU), so the PP-major consecutive-rank layout is
    always adjacent; this probe VERIFIES that instead of assuming it.
    Non-fatal: prints a warning on rank 0 if a stage edge lacks p2p.
    """
    if topo.pp <= 1 or topo.rank != 0:
        return
    try:
        ndev = torch.cuda.device_count()
        if ndev < 2:
            return  # single visible GPU: nothing to verify
        missing = []
        for s in range(topo.pp - 1):
            a, b = s % ndev, (s + 1) % ndev
            if a != b and not torch.cuda.can_device_access_peer(a, b):
                missing.append((a, b))
        if missing:
            print(f"[topology] WARNING: PP stage edges without GPU p2p "
                  f"(xGMI) access: {missing} — stage traffic will bounce "
                  f"through host memory", flush=True)
        else:
            print(f"[topology] PP stage edges peer-accessible (xGMI) "
                  f"across {min(topo.pp, ndev)} devices ✓", flush=True)
    except Exception as e:  # never block training on a probe
        print(f"[topology] probe skipped: {e}", flush=True)


# ---------------------------------------------------------------- p2p
#
# On the production path (one rank per GPU, backend "nccl" == RCCL on
# ROCm) these are direct device-to-device xGMI transfers.  The gloo
# backend cannot move CUDA tensors p2p, so a CPU staging fallback keeps
# the SAME Worker/schedule code testable with several ranks sharing one
# GPU (or none) — used by the multi-process GPU integration tests.


def _needs_cpu_staging(t: torch.Tensor) -> bool:
    return t.is_cuda and dist.get_backend() == "gloo"


class _StagedWork:
    """Work-handle wrapper keeping the staged CPU copy alive until
    wait() (isend fallback)."""

    def __init__(self, work, staged):
        self._work = work
        self._staged = staged

    def wait(self):
        self._work.wait()
        self._staged = None


def send_tensor(t: torch.Tensor, dst_rank: int):
    if _needs_cpu_staging(t):
        dist.send(t.detach().cpu(), dst=dst_rank)
        return
    dist.send(t.contiguous(), dst=dst_rank)


def isend_tensor(t: torch.Tensor, dst_rank: int):
    """Nonblocking send; caller waits the returned work handle before
    reusing the buffer (double-buffered PP edges)."""
    if _needs_cpu_staging(t):
        staged = t.detach().cpu()
        return _StagedWork(dist.isend(staged, dst=dst_rank), staged)
    return dist.isend(t.contiguous(), dst=dst_rank)


def recv_tensor(t: torch.Tensor, src_rank: int):
    if _needs_cpu_staging(t):
        tmp = torch.empty(t.shape, dtype=t.dtype, device="cpu")
        dist.recv(tmp, src=src_rank)
        t.copy_(tmp, non_blocking=True)
        return
    dist.recv(t, src=src_rank)


def can_batch_p2p(device=None) -> bool:
    """Batched p2p runs on the RCCL device path and on gloo with CPU
    tensors (so the multi-process CPU tests exercise the same batched
    code path the GPU uses).  gloo with CUDA tensors keeps the per-op
    CPU-staging fallback."""
    if not dist.is_initialized():
        return False
    backend = dist.get_backend()
    if backend == "nccl":
        return True
    return backend == "gloo" and device is not None and device.type == "cpu"


def batch_p2p(ops):
    """Issue a set of p2p ops as ONE grouped RCCL call.

    ops: list of ("send"|"recv", device_tensor, peer_rank).
    dist.batch_isend_irecv wraps them in ncclGroupStart/End, giving
    NCCL's matched-order semantics for simultaneous bidirectional stage
    edges (e.g. send-activations-to-next + recv-grad-from-next in one
    step) instead of relying on per-call enqueue order — the hardening
    recommended for first RCCL contact.  Returns the work handles; for
    NCCL wait() is a stream-level wait, not a host block.
    """
    p2p = [dist.P2POp(dist.isend if kind == "send" else dist.irecv, t, peer)
           for kind, t, peer in ops]
    return dist.batch_isend_irecv(p2p)


# ------------------------------------------------------- grad reduction

class GradReducer:
    """Bucketed, backward-overlapped DP gradient all-reduce.

    Buckets are contiguous slices of the stage's flat f32 grad buffer,
    grouped along layer boundaries in BACKWARD order (layers are laid
    out forward-contiguously, so a run of consecutive layers is one
    contiguous slice).  param_done(p) is called from the Sequential
    grad hook as each parameter's grad becomes final
    (pipe.py:302-316 semantics); when a bucket's last param arrives its
    async all-reduce launches.  finalize() waits on every in-flight
    handle (pipe.py:319-327 Waitall semantics) — call it from the
    post-grad hook, before OptimizerStep.
    """

    def __init__(self, model, group, bucket_bytes=None):
        self.group = group
        self.flat = model._flat_grad
        if bucket_bytes is None:
            # adaptive: ~4 buckets per stage so earlier buckets'
            # all-reduces overlap the remaining backward compute, but
            # never below 256 KB (latency-bound regime on xGMI) nor
            # above 25 MB (DDP-style cap, reference pipe.py:309-310)
            total = self.flat.numel() * 4
            bucket_bytes = max(256 << 10, min(25 << 20, total // 4))
        self.enabled = group is not None and dist.is_initialized() \
            and dist.get_world_size(group=group) > 1

        # layer -> (start, end) in flat buffer; params -> bucket id
        spans = []  # per layer in forward order
        off = 0
        layer_params = []
        for layer in model.layers:
            ps = [p for p in layer.parameters() if p.requires_grad]
            n = sum(p.data.numel() for p in ps)
            spans.append((off, off + n))
            layer_params.append(ps)
            off += n
        assert off == self.flat.numel()

        # group layers into buckets in backward (reverse) order
        self.buckets = []  # list of dicts
        self.param_bucket = {}
        cur_lo, cur_hi, cur_params, cur_bytes = None, None, [], 0
        for li in reversed(range(len(spans))):
            lo, hi = spans[li]
            nbytes = (hi - lo) * 4
            if cur_lo is not None and cur_bytes + nbytes > bucket_bytes:
                self._push_bucket(cur_lo, cur_hi, cur_params)
                cur_lo, cur_hi, cur_params, cur_bytes = None, None, [], 0
            if hi == lo:
                continue
            cur_hi = hi if cur_hi is None else cur_hi
            cur_lo = lo
            cur_params += layer_params[li]
            cur_bytes += nbytes
        if cur_lo is not None:
            self._push_bucket(cur_lo, cur_hi, cur_params)

        self._pending = {}
        self._handles = []
        self.reset()

    def _push_bucket(self, lo, hi, params):
        bid = len(self.buckets)
        self.buckets.append({"lo": lo, "hi": hi, "nparams": len(params)})
        for p in params:
            self.param_bucket[id(p)] = bid

    def reset(self):
        self._pending = {i: b["nparams"] for i, b in enumerate(self.buckets)}
        self._handles = []

    def param_done(self, p):
        if not self.enabled:
            return
        bid = self.param_bucket.get(id(p))
        if bid is None:
            return
        self._pending[bid] -= 1
        if self._pending[bid] == 0:
            b = self.buckets[bid]
            h = dist.all_reduce(
                self.flat[b["lo"]:b["hi"]], op=dist.ReduceOp.SUM,
                group=self.group, async_op=True,
            )
            self._handles.append(h)

    def finalize(self):
        if not self.enabled:
            return
        for h in self._handles:
            h.wait()
        # any bucket that never fired (e.g. a param whose hook did not
        # run) is a bug — fail loudly rather than silently desync.
        missed = [i for i, n in self._pending.items() if n > 0]
        if missed:
            raise RuntimeError(f"grad buckets never completed: {missed}")
        self.reset()
