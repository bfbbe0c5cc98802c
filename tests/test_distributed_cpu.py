"""Multi-process CPU tests (gloo backend, world_size 2): DP replica
sync + serial-equivalence, PP stage equivalence — the distributed
correctness gates the reference verifies via hash-sync and
serial-vs-distributed comparison (train.py:154-155,
scripts/DDP_PyTorch_MNIST.py:159-167)."""

import os
import random

import pytest
import torch
import torch.multiprocessing as mp

from shallowspeed_amd.data import Dataset
from shallowspeed_amd.models import MLP, SGD
from shallowspeed_amd.parallel import SCHEDULES, NaiveParallelSchedule, Topology, Worker
from shallowspeed_amd.utils import assert_sync, get_model_hash

SIZES = [24, 16, 12, 8, 6, 10]
GBS = 32
MUB = 4
N = 64
LR = 0.05
STEPS_BATCHES = 2


def _serial_params(loss="mse", schedule="naive", sizes=None):
    sizes = sizes or SIZES
    model = MLP(sizes, 0, 1, GBS, loss=loss).materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, MUB, n_samples=N, in_dim=sizes[0], n_classes=sizes[-1])
    ds.load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    return [p.data.clone() for p in model.parameters()]


def _dist_entry(rank, world, port, fn_name, out_dir, kwargs):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world)
    fn = globals()[fn_name]
    fn(rank, world, out_dir, **kwargs)


def _run_dist(fn, world, tmp_path, **kwargs):
    port = random.randint(20000, 45000)
    mp.spawn(
        _dist_entry,
        args=(world, port, fn.__name__, str(tmp_path), kwargs),
        nprocs=world,
        join=True,
    )


# ------------------------------------------------------------------ DP

def _dp_train(rank, world, out_dir, loss="mse", schedule="naive",
              bucket_bytes=25 << 20, force_defer=False, optimizer="sgd"):
    from shallowspeed_amd.parallel import init_topology

    topo = init_topology(dp=world, pp=1, backend="gloo",
                         device=torch.device("cpu"))
    model = MLP(SIZES, 0, 1, GBS, loss=loss).materialize_device("cpu")
    if optimizer == "adamw":
        from shallowspeed_amd.models import AdamW

        opt = AdamW(model.parameters(), lr=LR / 5, weight_decay=0.01)
    else:
        opt = SGD(model.parameters(), lr=LR)
    # local batch = GBS/world, split into MUB µbatches per rank
    ds = Dataset(GBS, (GBS // world) // MUB,
                 n_samples=N, in_dim=SIZES[0], n_classes=SIZES[-1])
    ds.load(topo.dp_rank, world)
    w = Worker(topo, model, ds, opt, bucket_bytes=bucket_bytes)
    if bucket_bytes < (1 << 20):
        assert len(w.reducer.buckets) > 1, "test wants multiple buckets"
    seen = {"in_bwd": 0, "elsewhere": 0}
    if force_defer:
        # exercise the deferred-wgrad path (normally GPU-only) and
        # assert the bucket all-reduce hooks fire MID-backward (the
        # restored reference pipe.py:302-316 overlap), never at the
        # optimizer-step flush
        w._force_defer = True
        orig = w.reducer.param_done

        def spy(p):
            key = "in_bwd" if model._flush_in_backward else "elsewhere"
            seen[key] += 1
            return orig(p)

        w.reducer.param_done = spy
    cls = SCHEDULES[schedule]
    for b in range(ds.num_batches()):
        w.execute(cls(ds.num_mubatches(), 1, 0), b)
    if force_defer:
        assert seen["in_bwd"] > 0 and seen["elsewhere"] == 0, seen
    # replicas must hash-identical (reference train.py:154-155)
    assert_sync(topo.dp_group, get_model_hash(model))
    if rank == 0:
        torch.save([p.data for p in model.parameters()],
                   os.path.join(out_dir, "dp_params.pt"))
    torch.distributed.destroy_process_group()


@pytest.mark.parametrize("schedule", ["naive", "gpipe", "pipedream"])
def test_dp2_matches_serial(tmp_path, schedule):
    _run_dist(_dp_train, 2, tmp_path, schedule=schedule)
    got = torch.load(tmp_path / "dp_params.pt", weights_only=False)
    want = _serial_params()
    assert len(got) == len(want)
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)


def test_dp2_multibucket_matches_serial(tmp_path):
    """Small bucket size => several overlapped all-reduces per backward
    (the bucketed path with >1 bucket)."""
    _run_dist(_dp_train, 2, tmp_path, schedule="gpipe", bucket_bytes=2048)
    got = torch.load(tmp_path / "dp_params.pt", weights_only=False)
    want = _serial_params()
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)


def test_dp2_adamw_matches_serial(tmp_path):
    """AdamW under DP: identical summed grads on every replica must
    produce identical moments and parameters (hash-sync holds for
    stateful optimizers too), and match serial AdamW training."""
    _run_dist(_dp_train, 2, tmp_path, schedule="gpipe", optimizer="adamw")
    got = torch.load(tmp_path / "dp_params.pt", weights_only=False)

    from shallowspeed_amd.models import AdamW
    from shallowspeed_amd.parallel import NaiveParallelSchedule

    model = MLP(SIZES, 0, 1, GBS, loss="mse").materialize_device("cpu")
    opt = AdamW(model.parameters(), lr=LR / 5, weight_decay=0.01)
    ds = Dataset(GBS, MUB, n_samples=N, in_dim=SIZES[0], n_classes=SIZES[-1])
    ds.load(0, 1)
    w = Worker(Topology(), model, ds, opt)
    for b in range(ds.num_batches()):
        w.execute(NaiveParallelSchedule(ds.num_mubatches(), 1, 0), b)
    want = [p.data for p in model.parameters()]
    for g, wv in zip(got, want):
        torch.testing.assert_close(g, wv, rtol=1e-4, atol=1e-5)


@pytest.mark.parametrize("schedule", ["naive", "gpipe", "pipedream"])
def test_dp2_deferred_wgrad_matches_serial(tmp_path, schedule):
    """Deferred-µbatch wgrad (the GPU hot path, forced on CPU): grads
    flush mid-final-backward, bucket all-reduces overlap the remaining
    backward, replicas stay in sync and match serial training."""
    _run_dist(_dp_train, 2, tmp_path, schedule=schedule,
              bucket_bytes=2048, force_defer=True)
    got = torch.load(tmp_path / "dp_params.pt", weights_only=False)
    want = _serial_params()
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)


# ------------------------------------------------------------------ PP

def _pp_train(rank, world, out_dir, schedule="gpipe", sizes=None):
    from shallowspeed_amd.parallel import init_topology

    sizes = sizes or SIZES
    topo = init_topology(dp=1, pp=world, backend="gloo",
                         device=torch.device("cpu"))
    model = MLP(sizes, topo.stage_id, world, GBS,
                loss="mse").materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, GBS // MUB, n_samples=N, in_dim=sizes[0],
                 n_classes=sizes[-1])
    ds.load(0, 1)
    w = Worker(topo, model, ds, opt)
    cls = SCHEDULES[schedule]
    for b in range(ds.num_batches()):
        w.execute(cls(ds.num_mubatches(), world, topo.stage_id), b)
    torch.save([p.data for p in model.parameters()],
               os.path.join(out_dir, f"pp_stage{topo.stage_id}.pt"))
    torch.distributed.destroy_process_group()


@pytest.mark.parametrize("schedule", ["naive", "gpipe", "pipedream"])
def test_pp2_matches_serial(tmp_path, schedule):
    _run_dist(_pp_train, 2, tmp_path, schedule=schedule)
    s0 = torch.load(tmp_path / "pp_stage0.pt", weights_only=False)
    s1 = torch.load(tmp_path / "pp_stage1.pt", weights_only=False)
    want = _serial_params()
    got = s0 + s1
    assert len(got) == len(want)
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)


def test_pp3_1f1b_matches_serial(tmp_path):
    _run_dist(_pp_train, 3, tmp_path, schedule="pipedream")
    got = []
    for s in range(3):
        got += torch.load(tmp_path / f"pp_stage{s}.pt", weights_only=False)
    want = _serial_params()
    assert len(got) == len(want)
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)


# ------------------------------------------------------------ DP × PP

def _grid_train(rank, world, out_dir, dp=2, pp=2, schedule="gpipe"):
    from shallowspeed_amd.parallel import init_topology

    topo = init_topology(dp=dp, pp=pp, backend="gloo",
                         device=torch.device("cpu"))
    model = MLP(SIZES, topo.stage_id, pp, GBS,
                loss="mse").materialize_device("cpu")
    opt = SGD(model.parameters(), lr=LR)
    ds = Dataset(GBS, (GBS // dp) // MUB, n_samples=N, in_dim=SIZES[0],
                 n_classes=SIZES[-1])
    ds.load(topo.dp_rank, dp)
    w = Worker(topo, model, ds, opt)
    cls = SCHEDULES[schedule]
    for b in range(ds.num_batches()):
        w.execute(cls(ds.num_mubatches(), pp, topo.stage_id), b)
    assert_sync(topo.dp_group, get_model_hash(model))
    if topo.dp_rank == 0:
        torch.save([p.data for p in model.parameters()],
                   os.path.join(out_dir, f"grid_stage{topo.stage_id}.pt"))
    torch.distributed.destroy_process_group()


def test_dp2xpp2_matches_serial(tmp_path):
    _run_dist(_grid_train, 4, tmp_path, dp=2, pp=2, schedule="gpipe")
    got = []
    for s in range(2):
        got += torch.load(tmp_path / f"grid_stage{s}.pt", weights_only=False)
    want = _serial_params()
    assert len(got) == len(want)
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)


def test_pp4_1f1b_matches_serial(tmp_path):
    """Deepest CPU-testable pipeline: 4 stages, 1F1B, world 4
    (8 layer boundaries so len(sizes) divides into 4 stages)."""
    sizes8 = [24, 16, 12, 8, 6, 12, 8, 10]
    _run_dist(_pp_train, 4, tmp_path, schedule="pipedream", sizes=sizes8)
    got = []
    for s in range(4):
        got += torch.load(tmp_path / f"pp_stage{s}.pt", weights_only=False)
    want = _serial_params(sizes=sizes8)
    assert len(got) == len(want)
    for g, w in zip(got, want):
        torch.testing.assert_close(g, w, rtol=1e-4, atol=1e-5)
